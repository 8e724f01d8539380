"""Pipeline parallelism: layer-range stages + 1F1B schedule on P2P sends.

MI355X-native rebuild of the capability the reference reached through
vendored ColossalAI (colossalai/pipeline/p2p.py:153-260 batched P2P,
schedule/one_f_one_b.py 1F1B). Design:

- each PP rank keeps the layer range it owns (stage 0 adds embeddings, the
  last stage adds final-norm + lm_head + loss) and FREES the rest of the
  model, so stage memory is model_size / pp;
- the schedule is 1F1B: (pp - rank - 1) warmup forwards, then alternating
  forward/backward at depth 1, then cooldown backwards — peak in-flight
  activations per stage = pp - rank;
- boundary tensors are fixed-shape [micro, S, h] bf16 over dist.send/recv
  (RCCL P2P rides a dedicated xGMI link between adjacent GPUs; gloo in CPU
  tests);
- MoE aux losses of intermediate stages enter each stage's local backward
  directly (torch.autograd.backward([y, aux], [gy, 1])), so load-balance
  gradients do not need to travel with the loss.

Composition: PP is standalone in this version (no PP x DP/EP mesh yet).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn

from . import comm


class PipelineStage(nn.Module):
    """The slice of a DeepSeekTransformer owned by one PP rank."""

    def __init__(self, model, lo: int, hi: int, is_first: bool, is_last: bool):
        super().__init__()
        self.is_first = is_first
        self.is_last = is_last
        self.rotary = model.rotary
        self.embed_scale = model.embed_scale
        self.embed_tokens = model.embed_tokens if is_first else None
        self.layers = nn.ModuleList(model.layers[lo:hi])
        self.final_norm = model.final_norm if is_last else None
        self.lm_head = model.lm_head if is_last else None
        # tied embeddings on a multi-stage pipeline must be untied (the
        # reference's tying assumes co-located embed/head)
        if is_last and not is_first and model.config.tie_word_embeddings:
            self.lm_head = nn.Linear(model.config.hidden_size,
                                     model.config.vocab_size, bias=False)
            with torch.no_grad():
                self.lm_head.weight.copy_(model.lm_head.weight)

    def forward(self, x: torch.Tensor):
        """x: token ids (first stage) or hidden states. Returns
        (output, aux_loss_sum)."""
        if self.is_first:
            x = self.embed_tokens(x) * self.embed_scale
        S = x.shape[1]
        rope_cs = self.rotary.get(S, x.device)
        aux_total = x.new_zeros(())
        for layer in self.layers:
            x, aux = layer(x, rope_cs)
            if layer.is_moe:
                aux_total = aux_total + aux
        if self.is_last:
            x = self.final_norm(x)
            x = self.lm_head(x)
        return x, aux_total


def partition_layers(num_layers: int, pp: int) -> List[Tuple[int, int]]:
    """Contiguous near-even split; earlier stages get the remainder (they
    also hold the embedding)."""
    base = num_layers // pp
    rem = num_layers % pp
    bounds = []
    lo = 0
    for s in range(pp):
        hi = lo + base + (1 if s < rem else 0)
        bounds.append((lo, hi))
        lo = hi
    return bounds


class PipelineParallelEngine:
    """1F1B pipelined training over a PP process group."""

    def __init__(self, model, config, pp_group=None,
                 loss_fn=None, device: Optional[torch.device] = None):
        self.pg = pp_group
        self.pp = dist.get_world_size(pp_group) if dist.is_initialized() else 1
        self.rank = dist.get_rank(pp_group) if dist.is_initialized() else 0
        self.device = device or next(model.parameters()).device
        self.hidden = model.config.hidden_size
        bounds = partition_layers(model.config.num_layers, self.pp)
        lo, hi = bounds[self.rank]
        self.stage = PipelineStage(model, lo, hi,
                                   is_first=self.rank == 0,
                                   is_last=self.rank == self.pp - 1)
        self.loss_fn = loss_fn
        self._dtype = next(self.stage.parameters()).dtype

    # ------------------------------------------------------------- p2p
    def _send(self, t: torch.Tensor, dst_stage: int):
        """Non-blocking send: 1F1B's steady state has adjacent stages
        sending to each other simultaneously (activation down, grad up) —
        blocking rendezvous sends would deadlock there."""
        t = t.contiguous()
        w = dist.isend(t, self._global(dst_stage), group=self.pg)
        self._inflight.append((w, t))   # keep the buffer alive until waited

    def _drain_sends(self):
        for w, _ in self._inflight:
            w.wait()
        self._inflight.clear()

    def _global(self, stage_rank: int) -> int:
        if self.pg is None:
            return stage_rank
        return dist.get_process_group_ranks(self.pg)[stage_rank]

    # ------------------------------------------------------------- 1F1B
    def train_batch(self, micro_batches: List[Dict]) -> Dict:
        """Run one optimizer-step's micro-batches through 1F1B.
        micro_batches: list of dicts with input_ids/labels(/loss_weights);
        every stage receives the SAME list (only the fields it needs are
        used). Returns {'loss': mean CE (last stage; zeros elsewhere)}."""
        n = len(micro_batches)
        pp, r = self.pp, self.rank
        self._inflight = []
        warmup = min(pp - r - 1, n)
        fwd_q: List[Tuple[torch.Tensor, torch.Tensor, torch.Tensor]] = []
        losses = []
        B = micro_batches[0]["input_ids"].shape[0]
        S = micro_batches[0]["input_ids"].shape[1]
        shape = (B, S, self.hidden)

        fwd_i = 0
        bwd_i = 0

        def forward_one():
            nonlocal fwd_i
            mb = micro_batches[fwd_i]
            if r == 0:
                x_in = mb["input_ids"].to(self.device)
            else:
                x_in = torch.empty(shape, dtype=self._dtype,
                                   device=self.device)
                dist.recv(x_in, self._global(r - 1), group=self.pg)
                x_in.requires_grad_(True)
            y, aux = self.stage(x_in)
            if r == pp - 1:
                labels = mb["labels"].to(self.device)
                w = mb.get("loss_weights")
                if w is not None:
                    w = w.to(self.device)
                from ..ops import fused_cross_entropy
                ce, acc, nv = fused_cross_entropy(y, labels, w)
                losses.append(ce.detach())
                fwd_q.append((x_in, ce + aux, None))
            else:
                self._send(y, r + 1)
                fwd_q.append((x_in, y, aux))
            fwd_i += 1

        def backward_one():
            nonlocal bwd_i
            x_in, out, aux = fwd_q.pop(0)
            if r == pp - 1:
                out.backward()          # out == ce + aux (scalar)
            else:
                gy = torch.empty(shape, dtype=self._dtype, device=self.device)
                dist.recv(gy, self._global(r + 1), group=self.pg)
                if aux is not None and aux.requires_grad:
                    torch.autograd.backward([out, aux],
                                            [gy, torch.ones_like(aux)])
                else:
                    out.backward(gy)
            if r > 0:
                self._send(x_in.grad, r - 1)
            bwd_i += 1

        for _ in range(warmup):
            forward_one()
        while fwd_i < n:
            forward_one()
            backward_one()
        while bwd_i < n:
            backward_one()
        self._drain_sends()

        mean_loss = (torch.stack(losses).mean() if losses
                     else torch.zeros((), device=self.device))
        return {"loss": mean_loss, "n_micro": n}

    def parameters(self):
        return self.stage.parameters()


class InterleavedPipelineEngine:
    """Interleaved virtual-stage pipeline (reference capability:
    vendored colossalai/pipeline/schedule/interleaved_pp.py).

    The model is split into pp x v contiguous chunks; rank r owns the v
    chunks with global stage index s = c*pp + r, so each micro-batch hops
    between ranks v times and the per-hop bubble shrinks by ~v. The
    schedule here is breadth-first (all forwards in global stage order,
    then all backwards in reverse) — simple, deadlock-free on blocking
    recv + isend, GPipe-like activation memory. Depth-first 1F1B stays the
    default for v=1 (PipelineParallelEngine).
    """

    def __init__(self, model, config, virtual_stages: int = 2,
                 pp_group=None, loss_fn=None,
                 device: Optional[torch.device] = None):
        self.pg = pp_group
        self.pp = dist.get_world_size(pp_group) if dist.is_initialized() else 1
        self.rank = dist.get_rank(pp_group) if dist.is_initialized() else 0
        self.v = max(1, int(virtual_stages))
        self.device = device or next(model.parameters()).device
        self.hidden = model.config.hidden_size
        total = self.pp * self.v
        if model.config.num_layers < total:
            raise ValueError(f"need >= {total} layers for pp={self.pp} x "
                             f"v={self.v}")
        bounds = partition_layers(model.config.num_layers, total)
        self.last_stage = total - 1
        self.chunks = nn.ModuleList()
        self.stage_ids: List[int] = []
        for c in range(self.v):
            s = c * self.pp + self.rank
            lo, hi = bounds[s]
            self.chunks.append(PipelineStage(
                model, lo, hi, is_first=s == 0, is_last=s == self.last_stage))
            self.stage_ids.append(s)
        self._dtype = next(self.chunks.parameters()).dtype

    def _global(self, stage_rank: int) -> int:
        if self.pg is None:
            return stage_rank
        return dist.get_process_group_ranks(self.pg)[stage_rank]

    def _send(self, t: torch.Tensor, dst_stage_rank: int):
        t = t.contiguous()
        w = dist.isend(t, self._global(dst_stage_rank), group=self.pg)
        self._inflight.append((w, t))

    def train_batch(self, micro_batches: List[Dict]) -> Dict:
        n = len(micro_batches)
        pp, v = self.pp, self.v
        self._inflight = []
        B, S = micro_batches[0]["input_ids"].shape
        shape = (B, S, self.hidden)
        losses = []
        # acts[c][i] = (x_in, out, aux); for pp==1 chunks chain locally
        acts: List[List[Tuple]] = [[None] * n for _ in range(v)]
        local_y: List[List[Optional[torch.Tensor]]] = \
            [[None] * n for _ in range(v)]

        # ---- forward sweep, global stage order
        for c in range(v):
            s = self.stage_ids[c]
            for i in range(n):
                if s == 0:
                    x_in = micro_batches[i]["input_ids"].to(self.device)
                elif pp == 1:
                    x_in = local_y[c - 1][i].detach().requires_grad_(True)
                else:
                    x_in = torch.empty(shape, dtype=self._dtype,
                                       device=self.device)
                    dist.recv(x_in, self._global((s - 1) % pp), group=self.pg)
                    x_in.requires_grad_(True)
                y, aux = self.chunks[c](x_in)
                if s == self.last_stage:
                    mb = micro_batches[i]
                    labels = mb["labels"].to(self.device)
                    w = mb.get("loss_weights")
                    if w is not None:
                        w = w.to(self.device)
                    from ..ops import fused_cross_entropy
                    ce, _, _ = fused_cross_entropy(y, labels, w)
                    losses.append(ce.detach())
                    acts[c][i] = (x_in, ce + aux, None)
                else:
                    if pp == 1:
                        local_y[c][i] = y
                    else:
                        self._send(y, (s + 1) % pp)
                    acts[c][i] = (x_in, y, aux)

        # ---- backward sweep, reverse global stage order
        local_gy: List[List[Optional[torch.Tensor]]] = \
            [[None] * n for _ in range(v)]
        for c in reversed(range(v)):
            s = self.stage_ids[c]
            for i in range(n):
                x_in, out, aux = acts[c][i]
                if s == self.last_stage:
                    out.backward()
                else:
                    if pp == 1:
                        gy = local_gy[c + 1][i]
                    else:
                        gy = torch.empty(shape, dtype=self._dtype,
                                         device=self.device)
                        dist.recv(gy, self._global((s + 1) % pp),
                                  group=self.pg)
                    if aux is not None and aux.requires_grad:
                        torch.autograd.backward([out, aux],
                                                [gy, torch.ones_like(aux)])
                    else:
                        out.backward(gy)
                if s > 0:
                    if pp == 1:
                        local_gy[c][i] = x_in.grad
                    else:
                        self._send(x_in.grad, (s - 1) % pp)
                acts[c][i] = None
        for w_, _ in self._inflight:
            w_.wait()
        self._inflight.clear()

        mean_loss = (torch.stack(losses).mean() if losses
                     else torch.zeros((), device=self.device))
        return {"loss": mean_loss, "n_micro": n}

    def parameters(self):
        return self.chunks.parameters()
