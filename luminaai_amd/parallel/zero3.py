"""ZeRO-3: parameter + gradient + optimizer-state sharding on RCCL/xGMI.

Replaces the reference's DeepSpeed ZeRO-3 / FSDP FULL_SHARD / ColossalAI
Gemini paths (reference backend_deepspeed.py:129-165, backend_fsdp.py:151-198,
vendored zero/gemini) with one engine built for this framework's flat-buffer
model:

- sharding unit = one TransformerBlock (plus a root unit: embeddings /
  final norm / untied head) — each unit's params live as views into a
  RESIZABLE flat bf16 buffer; only the rank's 1/world shard (bf16 weight +
  fp32 master/m/v + bf16 grad accumulator) persists.
- forward: a pre-hook all-gathers the unit's flat buffer from the bf16 weight
  shards (prefetching the next unit on a side stream), a post-hook frees the
  full buffer (storage resize-to-0, so autograd-saved views stay valid).
- backward: weights re-materialise either via the module's
  full_backward_pre_hook (plain path) or via the forward pre-hook firing
  inside activation-recompute (checkpointed path — recompute is detected with
  torch._C._current_graph_task_id(), which is only >= 0 inside a backward
  pass, so the free-after-forward hook knows not to free mid-recompute).
- after a unit's last param grad accumulates, the full grad buffer is
  reduce-scattered into the persistent shard accumulator and freed; gradient
  accumulation over micro-batches sums in the shard (same bf16-accumulate
  semantics as the ZeRO-1/2 engine).
- optimizer: the fused HIP clip+AdamW kernel updates each shard in place
  (master fp32 -> bf16 weight shard); there is NO post-step weight
  all-gather — the next forward's unit gathers are the re-materialisation.

xGMI sizing: unit buffers for the b-series blocks are 30-300 MB — each
all-gather is one fully-connected exchange over the 7 links; the per-unit
pipeline (prefetch next while computing current) keeps links and CUs busy.
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from ..ops import interface as K
from . import comm

ALIGN = 256


def _in_backward() -> bool:
    """True inside a backward pass (including checkpoint recompute)."""
    try:
        return torch._C._current_graph_task_id() != -1
    except AttributeError:
        return False


def _decay_split(named_params):
    from ..training.optimizer import NO_DECAY_KEYWORDS
    decay, no_decay = [], []
    for name, p in named_params:
        if not p.requires_grad:
            continue
        if any(k in name for k in NO_DECAY_KEYWORDS) or p.dim() <= 1:
            no_decay.append(p)
        else:
            decay.append(p)
    return decay, no_decay


class _Zero3Segment:
    """One (params, weight_decay) flat segment of a sharding unit."""

    def __init__(self, params: List[torch.Tensor], lr: float,
                 weight_decay: float, rank: int, world: int, pg=None,
                 comm_kind: str = "dp", offload: bool = False):
        self.params = params
        self.lr = lr
        self.weight_decay = weight_decay
        self.rank = rank
        self.world = world
        self.pg = pg
        self.comm = comm_kind
        self.numel = sum(p.numel() for p in params)
        q = ALIGN * world
        self.padded = (self.numel + q - 1) // q * q
        self.shard_size = self.padded // world
        lo = rank * self.shard_size
        hi = lo + self.shard_size

        device = params[0].device
        self.dtype = params[0].dtype
        self.flat_w = torch.zeros(self.padded, device=device, dtype=self.dtype)
        self.flat_g = torch.zeros(self.padded, device=device, dtype=self.dtype)
        self.offsets = []
        off = 0
        for p in params:
            n = p.numel()
            self.flat_w[off:off + n].copy_(p.data.reshape(-1).to(self.dtype))
            p.data = self.flat_w[off:off + n].view(p.shape)
            p.grad = self.flat_g[off:off + n].view(p.shape)
            self.offsets.append((off, n))
            off += n

        # persistent shard state.  offload=True keeps the fp32 master and
        # the Adam moments in host (pinned) memory and runs the update on
        # the CPU (the ZeRO-3 analogue of FlatAdamW's cpu_offload_optimizer
        # path; reference backend_deepspeed.py:129-165 offload_optimizer):
        # only the bf16 w_shard/g_shard stay resident, cutting per-rank
        # optimizer HBM from 12 to 4 bytes/param.
        self.offload = bool(offload)
        pin = device.type == "cuda"
        state_dev = torch.device("cpu") if self.offload else device
        self.w_shard = self.flat_w[lo:hi].clone()
        mast = self.w_shard.float()
        self.master = (mast.cpu().pin_memory() if self.offload and pin
                       else mast.cpu() if self.offload else mast.clone())
        self.g_shard = torch.zeros_like(self.w_shard)
        self.m = torch.zeros(self.shard_size, device=state_dev,
                             dtype=torch.float32)
        self.v = torch.zeros_like(self.m)
        if self.offload and pin:
            self._g_stage = torch.zeros(self.shard_size, dtype=self.dtype,
                                        pin_memory=True)
            self._w_stage = torch.zeros(self.shard_size, dtype=self.dtype,
                                        pin_memory=True)
        elif self.offload:
            # CPU device (tests): the staging buffers alias the shards so
            # the same two-phase step path runs without extra copies
            self._g_stage = self.g_shard
            self._w_stage = self.w_shard

    # -------------------------------------------------- storage management
    def weights_alloc(self) -> bool:
        return self.flat_w.untyped_storage().size() > 0

    def free_weights(self):
        self.flat_w.untyped_storage().resize_(0)

    def free_grads(self):
        self.flat_g.untyped_storage().resize_(0)

    def alloc_grads(self):
        st = self.flat_g.untyped_storage()
        if st.size() == 0:
            st.resize_(self.padded * self.flat_g.element_size())
        self.flat_g.zero_()

    def gather(self):
        pg = self.pg
        st = self.flat_w.untyped_storage()
        if st.size() > 0:
            return
        st.resize_(self.padded * self.flat_w.element_size())
        if self.world == 1:
            self.flat_w.copy_(self.w_shard)
        elif dist.get_backend(pg) == "gloo":
            chunks = list(self.flat_w.chunk(self.world))
            dist.all_gather(chunks, self.w_shard.contiguous(), group=pg)
        else:
            dist.all_gather_into_tensor(self.flat_w, self.w_shard, group=pg)

    def reduce_into_shard(self):
        """flat_g -> += g_shard (sum over ranks), then free flat_g."""
        pg = self.pg
        lo = self.rank * self.shard_size
        hi = lo + self.shard_size
        if self.world == 1:
            self.g_shard.add_(self.flat_g[lo:hi])
        elif dist.get_backend(pg) == "gloo":
            dist.all_reduce(self.flat_g, group=pg)
            self.g_shard.add_(self.flat_g[lo:hi])
        else:
            tmp = torch.empty_like(self.g_shard)
            dist.reduce_scatter_tensor(tmp, self.flat_g, group=pg)
            self.g_shard.add_(tmp)
        self.free_grads()

    def step(self, step_count: int, norm_sq, max_norm: float,
             grad_scale: float):
        assert not self.offload, "offloaded segments step via step_host()"
        K.adamw_step(self.master, self.g_shard, self.m, self.v, self.w_shard,
                     self.lr, 0.9, 0.95, 1e-8, self.weight_decay, step_count,
                     norm_sq, max_norm, grad_scale)

    def stage_grad_d2h(self):
        if self._g_stage is not self.g_shard:
            self._g_stage.copy_(self.g_shard, non_blocking=True)

    def step_host(self, step_count: int, norm_cpu, max_norm: float,
                  grad_scale: float):
        """CPU AdamW on the host-resident state (after stage_grad_d2h +
        a device sync); async H2D of the refreshed weight shard."""
        K.adamw_step(self.master, self._g_stage, self.m, self.v,
                     self._w_stage, self.lr, 0.9, 0.95, 1e-8,
                     self.weight_decay, step_count, norm_cpu, max_norm,
                     grad_scale)
        if self._w_stage is not self.w_shard:
            self.w_shard.copy_(self._w_stage, non_blocking=True)

    def refresh_from_shard(self):
        """w_shard -> flat_w (persistent units after an optimizer step)."""
        pg = self.pg
        if self.flat_w.untyped_storage().size() == 0:
            return
        if self.world == 1:
            lo = self.rank * self.shard_size
            self.flat_w[lo:lo + self.shard_size].copy_(self.w_shard)
        elif dist.get_backend(pg) == "gloo":
            chunks = list(self.flat_w.chunk(self.world))
            dist.all_gather(chunks, self.w_shard.contiguous(), group=pg)
        else:
            dist.all_gather_into_tensor(self.flat_w, self.w_shard, group=pg)

    def sync_shard_from_full(self):
        """flat_w -> w_shard/master (after load_state_dict wrote params)."""
        lo = self.rank * self.shard_size
        self.w_shard.copy_(self.flat_w[lo:lo + self.shard_size])
        self.master.copy_(self.w_shard.float().to(self.master.device))


class _Zero3Unit:
    """A module whose parameters gather/free together."""

    def __init__(self, name: str, module: nn.Module, lr: float,
                 weight_decay: float, rank: int, world: int,
                 persistent: bool = False, pg=None, mesh=None,
                 offload: bool = False):
        self.name = name
        self.module = module
        self.persistent = persistent
        named = list(module.named_parameters())
        expert = []
        if mesh is not None and mesh.ep_size > 1:
            from .expert_parallel import is_expert_param
            expert = [(n, p) for n, p in named
                      if is_expert_param(n)
                      or getattr(p, "_shard_parallel", False)]
            eids = {id(p) for _, p in expert}
            named = [(n, p) for n, p in named if id(p) not in eids]
        decay, no_decay = _decay_split(named)
        self.segments: List[_Zero3Segment] = []
        if decay:
            self.segments.append(_Zero3Segment(decay, lr, weight_decay,
                                               rank, world, pg=pg,
                                               offload=offload))
        if no_decay:
            self.segments.append(_Zero3Segment(no_decay, lr, 0.0, rank, world,
                                               pg=pg, offload=offload))
        if expert:
            # EP-sharded expert weights: shard only across the ranks that
            # hold the SAME experts (the expert replica group); with full EP
            # (ep == world) that group is this rank alone -> local state
            r_world = mesh.shard_replica_size
            r_pg = mesh.shard_replica_group if r_world > 1 else None
            r_rank = (dist.get_process_group_ranks(r_pg).index(
                dist.get_rank()) if r_pg is not None else 0)
            self.segments.append(_Zero3Segment(
                [p for _, p in expert], lr, weight_decay,
                r_rank, r_world, pg=r_pg, comm_kind="expert",
                offload=offload))
        self.n_params = sum(len(s.params) for s in self.segments)
        self._grads_pending = self.n_params

    def gather(self):
        # If a prefetch stream gathered this unit's weights, the storage is
        # already allocated and segment gather() early-returns -- but the
        # all-gather may still be in flight on that stream.  Make the
        # consuming stream wait on the recorded event and tell the caching
        # allocator the buffers are now used by this stream too.
        ev = getattr(self, "_prefetch_event", None)
        if ev is not None:
            self._prefetch_event = None
            cur = torch.cuda.current_stream()
            cur.wait_event(ev)
            for s in self.segments:
                if s.flat_w.is_cuda:
                    s.flat_w.record_stream(cur)
        for s in self.segments:
            s.gather()

    def free_weights(self):
        if self.persistent:
            return
        for s in self.segments:
            s.free_weights()

    def alloc_grads(self):
        for s in self.segments:
            if s.flat_g.untyped_storage().size() == 0:
                s.alloc_grads()

    def grads_allocated(self) -> bool:
        return all(s.flat_g.untyped_storage().size() > 0
                   for s in self.segments)


class Zero3Optimizer:
    """FlatAdamW-compatible surface over the shard segments (the Trainer,
    scheduler and checkpointing code only touch this interface)."""

    def __init__(self, engine: "Zero3Engine", max_grad_norm: float = 1.0):
        self.engine = engine
        self.max_grad_norm = max_grad_norm
        self.step_count = 0
        self.shard_rank = engine.rank
        self.shard_world = engine.world
        self._last_norm_sq: Optional[torch.Tensor] = None

    @property
    def groups(self):
        return [s for u in self.engine.units for s in u.segments]

    @property
    def param_groups(self):
        return self.groups

    def zero_grad(self, set_to_none: bool = False):
        for g in self.groups:
            g.g_shard.zero_()

    def local_grad_norm_sq(self, shard_only: bool = True) -> torch.Tensor:
        total = None
        for g in self.groups:
            ns = K.l2norm_sq(g.g_shard)
            total = ns if total is None else total + ns
        return total

    @torch.no_grad()
    def step(self, grad_scale: float = 1.0, closure=None,
             norm_sq: Optional[torch.Tensor] = None, shard_only: bool = True):
        from ..ops.interface import invalidate_pad_cache
        invalidate_pad_cache()    # weights change: padded-K grad_x cache
        self.step_count += 1
        if norm_sq is None and self.max_grad_norm > 0:
            norm_sq = self.local_grad_norm_sq()
        self._last_norm_sq = norm_sq
        offl = [g for g in self.groups if g.offload]
        for g in self.groups:
            if not g.offload:
                g.step(self.step_count, norm_sq, self.max_grad_norm,
                       grad_scale)
        if offl:
            # device-resident segments' kernels are in flight; overlap the
            # D2H grad staging with them, sync once, then host math
            for g in offl:
                g.stage_grad_d2h()
            if torch.cuda.is_available() and any(
                    g.w_shard.is_cuda for g in offl):
                torch.cuda.synchronize()
            norm_cpu = (norm_sq.sum().cpu() if norm_sq is not None else None)
            if norm_cpu is not None:
                gn = float(norm_cpu) ** 0.5 * grad_scale
                if not math.isfinite(gn):
                    return      # NaN/Inf grads: skip the host update
            for g in offl:
                g.step_host(self.step_count, norm_cpu, self.max_grad_norm,
                            grad_scale)

    def last_grad_norm(self) -> float:
        if self._last_norm_sq is None:
            return 0.0
        val = float(self._last_norm_sq.sum())
        return math.sqrt(val) if math.isfinite(val) and val >= 0 else float("nan")

    def state_dict(self) -> Dict:
        return {
            "step_count": self.step_count,
            "zero_stage": 3,
            "shard_rank": self.shard_rank,
            "shard_world": self.shard_world,
            "groups": [
                {"master": g.master.cpu(), "m": g.m.cpu(), "v": g.v.cpu(),
                 "lr": g.lr, "weight_decay": g.weight_decay, "numel": g.numel}
                for g in self.groups
            ],
        }

    def load_state_dict(self, sd: Dict):
        if sd.get("shard_world", 1) != self.shard_world:
            raise ValueError("ZeRO-3 resharding of checkpoints not supported "
                             f"(saved world={sd.get('shard_world')})")
        self.step_count = sd.get("step_count", 0)
        for g, gs in zip(self.groups, sd["groups"]):
            g.master.copy_(gs["master"].to(g.master.device))
            g.m.copy_(gs["m"].to(g.m.device))
            g.v.copy_(gs["v"].to(g.v.device))
            g.lr = gs.get("lr", g.lr)
            g.weight_decay = gs.get("weight_decay", g.weight_decay)
            g.w_shard.copy_(g.master.to(g.w_shard.dtype))

    def load_resharded(self, shard_sds: List[Dict]):
        """Elastic resume at a different world size: concatenate every
        saved rank's shard (contiguous slices of the padded flat buffer)
        and re-slice for this topology (mirrors FlatAdamW.load_resharded;
        the Trainer routes here when shard_world mismatches)."""
        if any(g.comm == "expert" for g in self.groups):
            raise ValueError("elastic resharding with expert parallelism is "
                             "not supported (expert placement changes)")
        saved_world = shard_sds[0].get("shard_world", 1)
        if len(shard_sds) != saved_world:
            raise ValueError(f"need all {saved_world} shards, got "
                             f"{len(shard_sds)}")
        self.step_count = shard_sds[0].get("step_count", 0)
        for gi, g in enumerate(self.groups):
            saved_numel = shard_sds[0]["groups"][gi]["numel"]
            if saved_numel != g.numel:
                raise ValueError(f"group {gi} size mismatch ({g.numel} vs "
                                 f"{saved_numel})")
            lo = g.rank * g.shard_size
            hi = lo + g.shard_size
            for key, dst in (("master", g.master), ("m", g.m), ("v", g.v)):
                full = torch.cat([sd["groups"][gi][key].float()
                                  for sd in shard_sds])
                buf = torch.zeros(g.padded, dtype=torch.float32)
                buf[:g.numel] = full[:g.numel]
                dst.copy_(buf[lo:hi].to(dst.device))
            g.lr = shard_sds[0]["groups"][gi].get("lr", g.lr)
            g.weight_decay = shard_sds[0]["groups"][gi].get(
                "weight_decay", g.weight_decay)
            g.w_shard.copy_(g.master.to(g.w_shard.dtype))

    def rebuild(self, model):
        raise RuntimeError("dynamic expert add/prune is not supported under "
                           "ZeRO-3 (param shards are fixed); use ZeRO-0/1/2")


class Zero3Engine:
    """ZeroEngine-compatible engine: set_sync / reduce_gradients / step /
    zero_grad / broadcast_parameters, plus the unit hook machinery."""

    stage = 3

    def __init__(self, model: nn.Module, config, process_group=None,
                 mesh=None):
        self.model = model
        self.pg = process_group
        self.mesh = mesh
        if mesh is not None and getattr(mesh, "tp_size", 1) > 1:
            raise ValueError(
                "ZeRO-3 does not compose with TP (the dense-segment "
                "collectives assume the world group is pure DP x EP); "
                "use ZeRO-0/1/2 with tensor parallelism")
        self.world = comm.get_world_size()
        self.rank = comm.get_rank()
        self.bucket_bytes = getattr(config, "reduce_bucket_size", 50_000_000)
        self.overlap = getattr(config, "overlap_comm", True)
        lr = config.learning_rate
        wd = config.weight_decay
        offload = bool(getattr(config, "cpu_offload_optimizer", False)
                       or getattr(config, "aggressive_cpu_offload", False))

        blocks = list(getattr(model, "layers", []))
        # root unit: everything not inside a block (embed/final-norm/head)
        self.units: List[_Zero3Unit] = []
        root_mod = _RootShell(model, blocks)
        self.units.append(_Zero3Unit("root", root_mod, lr, wd,
                                     self.rank, self.world, persistent=True,
                                     pg=self.pg, mesh=mesh,
                                     offload=offload))
        for i, b in enumerate(blocks):
            self.units.append(_Zero3Unit(f"block{i}", b, lr, wd,
                                         self.rank, self.world,
                                         pg=self.pg, mesh=mesh,
                                         offload=offload))
        self.optimizer = Zero3Optimizer(self)
        self._unit_of_param: Dict[int, _Zero3Unit] = {}
        self._hooks = []
        self._install_hooks()
        self.sync_enabled = True
        self._prefetch_stream = (torch.cuda.Stream()
                                 if torch.cuda.is_available() else None)
        if self.world == 1:
            self._finalize_init()
        self._finalized = self.world == 1

    # ---------------------------------------------------------------- init
    def _finalize_init(self):
        """Free non-persistent unit weights (first gather re-materialises)."""
        for u in self.units:
            u.free_weights()
        self._finalized = True

    def broadcast_parameters(self):
        """Initial weights to all replicas, then shard + free. Dense
        segments broadcast from global rank 0; expert segments only within
        their replica group (each EP rank keeps its own experts)."""
        if self.world > 1:
            for u in self.units:
                for s in u.segments:
                    s.gather()  # no-op pre-finalize (still allocated)
                    if s.comm == "expert":
                        if s.pg is not None:
                            src = dist.get_process_group_ranks(s.pg)[0]
                            dist.broadcast(s.flat_w, src=src, group=s.pg)
                    else:
                        dist.broadcast(s.flat_w, src=0, group=self.pg)
                    lo = s.rank * s.shard_size
                    s.w_shard.copy_(s.flat_w[lo:lo + s.shard_size])
                    s.master.copy_(s.w_shard.float())
        self._finalize_init()

    # ---------------------------------------------------------------- hooks
    def _install_hooks(self):
        for ui, unit in enumerate(self.units):
            if unit.persistent:
                for s in unit.segments:
                    for p in s.params:
                        self._unit_of_param[id(p)] = unit
                        self._hooks.append(p.register_post_accumulate_grad_hook(
                            self._make_grad_hook(unit)))
                continue
            self._hooks.append(unit.module.register_forward_pre_hook(
                self._make_fwd_pre(ui)))
            self._hooks.append(unit.module.register_forward_hook(
                self._make_fwd_post(ui)))
            self._hooks.append(unit.module.register_full_backward_pre_hook(
                self._make_bwd_pre(ui)))
            for s in unit.segments:
                for p in s.params:
                    self._unit_of_param[id(p)] = unit
                    self._hooks.append(p.register_post_accumulate_grad_hook(
                        self._make_grad_hook(unit)))

    def _make_fwd_pre(self, ui):
        def hook(module, args):
            unit = self.units[ui]
            if not self._finalized:
                return
            unit.gather()
            if _in_backward():
                # checkpoint recompute: backward follows immediately
                unit.alloc_grads()
            elif self._prefetch_stream is not None and ui + 1 < len(self.units):
                nxt = self.units[ui + 1]
                # The prefetch stream must not race ahead of the compute
                # stream's last use of the buffers it is about to refill.
                self._prefetch_stream.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(self._prefetch_stream):
                    nxt.gather()
                ev = torch.cuda.Event()
                ev.record(self._prefetch_stream)
                nxt._prefetch_event = ev
        return hook

    def _make_fwd_post(self, ui):
        def hook(module, args, output):
            if not self._finalized or _in_backward():
                return  # inside recompute: the unit backward still needs them
            unit = self.units[ui]
            if module.training:
                unit.free_weights()
            # eval/generation: keep gathered for reuse across decode steps
        return hook

    def _make_bwd_pre(self, ui):
        def hook(module, grad_output):
            unit = self.units[ui]
            unit.gather()
            unit.alloc_grads()
            # backward visits units in reverse order: prefetch ui-1
            if self._prefetch_stream is not None and ui > 0:
                prev = self.units[ui - 1]
                if not prev.persistent:
                    self._prefetch_stream.wait_stream(
                        torch.cuda.current_stream())
                    with torch.cuda.stream(self._prefetch_stream):
                        prev.gather()
                    ev = torch.cuda.Event()
                    ev.record(self._prefetch_stream)
                    prev._prefetch_event = ev
        return hook

    def _make_grad_hook(self, unit):
        def hook(param):
            unit._grads_pending -= 1
            if unit._grads_pending == 0:
                unit._grads_pending = unit.n_params
                for s in unit.segments:
                    s.reduce_into_shard()
                    if unit.persistent:
                        # persistent units have no backward-pre hook to
                        # re-allocate grads: keep the buffer live (zeroed)
                        s.alloc_grads()
                unit.free_weights()
        return hook

    # ------------------------------------------------------------- engine API
    def set_sync(self, enabled: bool):
        # ZeRO-3 must reduce every micro-batch (full grads cannot persist)
        self.sync_enabled = True

    def reduce_gradients(self):
        pass  # handled per-unit by the grad hooks

    def global_grad_norm_sq(self) -> Optional[torch.Tensor]:
        if self.optimizer.max_grad_norm <= 0:
            return None
        ns = self.optimizer.local_grad_norm_sq()
        if self.world > 1:
            dist.all_reduce(ns, group=self.pg)
        return ns

    def step(self, grad_scale: float = 1.0):
        # flush units whose backward did not touch every param (e.g. a gate
        # with zero routed tokens): their grad buffers are still allocated
        for u in self.units:
            partial = u._grads_pending != u.n_params
            for s in u.segments:
                if partial or (not u.persistent
                               and s.flat_g.untyped_storage().size() > 0):
                    s.reduce_into_shard()
                    if u.persistent:
                        s.alloc_grads()
            u._grads_pending = u.n_params
            u.free_weights()
        norm_sq = self.global_grad_norm_sq()
        self.optimizer.step(grad_scale=grad_scale, norm_sq=norm_sq)
        for u in self.units:
            if u.persistent:
                for s in u.segments:
                    s.refresh_from_shard()

    def sync_shards_from_full(self):
        """After load_state_dict materialised full weights into flat buffers."""
        for u in self.units:
            for s in u.segments:
                s.sync_shard_from_full()

    def zero_grad(self):
        self.optimizer.zero_grad()

    def remove_hooks(self):
        for h in self._hooks:
            h.remove()
        self._hooks.clear()

    # ------------------------------------------------------------- checkpoint
    class _GatherAll:
        def __init__(self, engine):
            self.engine = engine

        def __enter__(self):
            for u in self.engine.units:
                u.gather()
            return self

        def __exit__(self, *exc):
            if self.engine.model.training:
                for u in self.engine.units:
                    u.free_weights()

    def gathered_weights(self):
        """Context manager materialising ALL params (state_dict/checkpoint)."""
        return Zero3Engine._GatherAll(self)


class _RootShell(nn.Module):
    """Wraps the model's non-block parameters as one pseudo-module so the
    root unit can flatten them (embeddings, final norm, untied lm_head)."""

    def __init__(self, model: nn.Module, blocks):
        super().__init__()
        in_blocks = set()
        for b in blocks:
            for p in b.parameters():
                in_blocks.add(id(p))
        self._names = []
        self._params = []
        for n, p in model.named_parameters():
            if id(p) not in in_blocks:
                self._names.append(n)
                self._params.append(p)

    def named_parameters(self, *a, **kw):
        return list(zip(self._names, self._params))

    def parameters(self, *a, **kw):
        return list(self._params)
