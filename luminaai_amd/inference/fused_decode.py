"""Fused batch-1 decoder: five weight-streaming kernels per layer.

The round-1 decode path ran ~18 small kernels per layer and measured 160
tok/s, latency-bound (profiles/r01_SUMMARY.md).  This decoder drives the
hand-written decode kernel family (ops/csrc/decode.hip) -- fused
norm+GEMV, RoPE+cache append, single-token GQA attention, fused
SwiGLU -- in the launches-baseline shape from the hardware guide, and
hipGraph-captures the whole token step (all shapes static, cursor on
device).

Scope: dense, MoD (MoD layers run dense at decode: the block's run_mod
gate requires S > 1) and single-GPU MoE models.  MoE layers decode as:
fused router kernel (rmsnorm + gate GEMV + on-device top-k) -> per selected
expert an indirect SWIGLU GEMV + scaled-residual down GEMV, the expert
index read from device memory so the step still hipGraph-captures.
Router softmax runs over bf16 logits (training gating is fp32): top-k
selection can differ from eager only on near-exact routing ties.
MoE with EP/TP shards, a load-balancer placement, or fp8/int8 expert
storage falls back to the standard engine path.
"""

from __future__ import annotations

from typing import List, Optional

import torch

from .. import ops

_NORM = 1
_RESID = 2
_SWIGLU = 4
_SCALE = 8


def can_fuse_decode(model) -> bool:
    if not (torch.cuda.is_available() and ops.has_ext()):
        return False
    e = ops.get_ext()
    if not hasattr(e, "dec_gemv"):
        return False
    for layer in model.layers:
        if getattr(layer, "is_moe", False):
            f = layer.ffn
            if (f.ep_size > 1 or getattr(f, "tp_group", None) is not None
                    or getattr(f, "placement", None) is not None
                    or getattr(f, "use_fp8", False)
                    or getattr(f, "use_int8_weights", False)
                    or f.num_experts > 64):
                return False
        a = layer.attention
        if a.head_dim > 192 or a.qkv_proj.weight.dtype != torch.bfloat16:
            return False
    return True


class FusedDecoder:
    """Greedy/sampled decoding on the fused kernel path.

    Usage:
        dec = FusedDecoder(model, max_context)
        logits = dec.prefill(ids)      # eager prompt pass (fills caches)
        logits = dec.step(token)       # fused single-token step
    """

    def __init__(self, model, max_context: int):
        assert can_fuse_decode(model)
        self.model = model
        self.max_context = max_context
        self.device = next(model.parameters()).device
        self.ext = ops.get_ext()
        h = model.layers[0].hidden_size
        a0 = model.layers[0].attention
        self.H, self.HKV, self.D = a0.num_heads, a0.num_kv_heads, a0.head_dim
        self.h = h
        self.eps = model.final_norm.eps
        self.scale = self.D ** -0.5

        # static work buffers
        dt, dev = torch.bfloat16, self.device
        qs, kvs = self.H * self.D, self.HKV * self.D
        self.x = torch.zeros(h, dtype=dt, device=dev)
        self.qkv = torch.zeros(qs + 2 * kvs, dtype=dt, device=dev)
        self.q = torch.zeros(qs, dtype=dt, device=dev)
        self.attn_out = torch.zeros(qs, dtype=dt, device=dev)
        imax = max(l.ffn.intermediate_size for l in model.layers)
        self.act = torch.zeros(imax, dtype=dt, device=dev)
        self.xhat = torch.zeros(h, dtype=dt, device=dev)
        self.x2 = torch.zeros(h, dtype=dt, device=dev)
        V = model.lm_head.weight.shape[0]
        self.logits = torch.zeros(V, dtype=dt, device=dev)
        self.pos_dev = torch.zeros(1, dtype=torch.int32, device=dev)
        self._in_tok = torch.zeros(1, dtype=torch.long, device=dev)
        self._len = 0                      # host mirror of pos_dev

        # RoPE tables (fp32 [S, D/2])
        cos, sin = model.rotary.get(max_context, dev)
        self.cos, self.sin = cos.contiguous(), sin.contiguous()

        # per-layer KV caches [cap, HKV, D]
        self.kc = [torch.zeros(max_context, self.HKV, self.D, dtype=dt,
                               device=dev) for _ in model.layers]
        self.vc = [torch.zeros_like(self.kc[0]) for _ in model.layers]

        # contiguous bf16 weights views; MoE experts pre-transposed into
        # GEMV row-major layout ([E, 2I, h] / [E, h, I]) once at init
        self.Wqkv, self.Wo, self.Wgu, self.Wdn = [], [], [], []
        self.wn_in, self.wn_post = [], []
        self.moe = []          # per-layer: None or (Wg, WguT, WdnT, k, temp)
        for layer in model.layers:
            at = layer.attention
            self.Wqkv.append(at.qkv_proj.weight.data.contiguous())
            self.Wo.append(at.o_proj.weight.data.contiguous())
            self.wn_in.append(layer.input_norm.weight.data.contiguous())
            self.wn_post.append(layer.post_attn_norm.weight.data.contiguous())
            if getattr(layer, "is_moe", False):
                f = layer.ffn
                self.Wgu.append(None)
                self.Wdn.append(None)
                self.moe.append((
                    f.gate.weight.data.to(dt).contiguous(),
                    f.w_gate_up.data.to(dt).permute(0, 2, 1).contiguous(),
                    f.w_down.data.to(dt).permute(0, 2, 1).contiguous(),
                    f.top_k, f.routing_temperature, f.intermediate_size))
            else:
                self.Wgu.append(
                    layer.ffn.gate_up_proj.weight.data.contiguous())
                self.Wdn.append(layer.ffn.down_proj.weight.data.contiguous())
                self.moe.append(None)
        kmax = max((m[3] for m in self.moe if m), default=0)
        self.eidx = torch.zeros(max(kmax, 1), dtype=torch.int32, device=dev)
        self.ew = torch.zeros(max(kmax, 1), dtype=torch.float32, device=dev)
        self.wn_final = model.final_norm.weight.data.contiguous()
        self.Wlm = model.lm_head.weight.data.contiguous()
        self.embed = model.embed_tokens.weight.data
        self.embed_scale = model.embed_scale
        self._graph: Optional[torch.cuda.CUDAGraph] = None

    @property
    def seq_len(self) -> int:
        return self._len                   # host mirror: no device sync

    def reset(self):
        self.pos_dev.zero_()
        self._len = 0

    @torch.no_grad()
    def prefill(self, ids: torch.Tensor) -> torch.Tensor:
        """Eager prompt pass through the model; copies its caches into the
        fused layout ([cap, HKV, D])."""
        from ..models.transformer import KVCache
        caches = [KVCache(max_len=self.max_context) for _ in self.model.layers]
        logits, _, _ = self.model(ids, kv_caches=caches)
        n = caches[0].seq_len
        for i, c in enumerate(caches):
            self.kc[i][:n] = c.k[0, :n]
            self.vc[i][:n] = c.v[0, :n]
        self.pos_dev.fill_(n)
        self._len = n
        return logits[:, -1]

    @torch.no_grad()
    def _one_step(self):
        e = self.ext
        x = self.x
        for i in range(len(self.model.layers)):
            e.dec_gemv(self.Wqkv[i], x, self.wn_in[i], None, self.qkv,
                       self.eps, _NORM)
            e.dec_rope_cache(self.qkv, self.q, self.kc[i], self.vc[i],
                             self.cos, self.sin, self.pos_dev,
                             self.H, self.HKV, self.D)
            e.dec_attn(self.q, self.kc[i], self.vc[i], self.attn_out,
                       self.pos_dev, self.H, self.HKV, self.D, self.scale)
            e.dec_gemv(self.Wo[i], self.attn_out, None, x, self.x2,
                       self.eps, _RESID)
            if self.moe[i] is None:
                e.dec_gemv(self.Wgu[i], self.x2, self.wn_post[i], None,
                           self.act, self.eps, _NORM | _SWIGLU)
                e.dec_gemv(self.Wdn[i], self.act, None, self.x2, x,
                           self.eps, _RESID)
            else:
                Wg, WguT, WdnT, k, temp, I = self.moe[i]
                e.dec_router(self.x2, self.wn_post[i], Wg, self.xhat,
                             self.eidx, self.ew, k, temp, self.eps)
                act = self.act[:I]
                for slot in range(k):
                    e.dec_gemv_moe(WguT, self.xhat, None, act,
                                   self.eidx, self.ew, slot, _SWIGLU)
                    e.dec_gemv_moe(WdnT, act,
                                   self.x2 if slot == 0 else x, x,
                                   self.eidx, self.ew, slot,
                                   _RESID | _SCALE)
        e.dec_gemv(self.Wlm, x, self.wn_final, None, self.logits,
                   self.eps, _NORM)
        e.dec_advance(self.pos_dev)

    @torch.no_grad()
    def _embed_in(self):
        # index_select keeps the token index on-device (int indexing would
        # sync and is hipGraph-capture-unsafe)
        row = torch.index_select(self.embed, 0, self._in_tok)
        self.x.copy_((row[0] * self.embed_scale).to(self.x.dtype))

    @torch.no_grad()
    def step(self, token_id: torch.Tensor) -> torch.Tensor:
        """token_id [1] long on device -> logits [V] (bf16)."""
        if self.seq_len >= self.max_context:
            raise RuntimeError("KV cache full")
        self._in_tok.copy_(token_id.view(1))
        self._len += 1
        if self._graph is not None:
            self._graph.replay()
            return self.logits
        self._embed_in()
        self._one_step()
        return self.logits

    @torch.no_grad()
    def capture(self):
        """hipGraph-capture the token step (shapes/cursor all device-side)."""
        if self._graph is not None or self.seq_len >= self.max_context:
            return
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            self._embed_in()
            self._one_step()
            self.pos_dev.sub_(1)          # undo the warmup advance
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._embed_in()
            self._one_step()
        # capture records but does not execute: the cursor did not move
        self._graph = g
