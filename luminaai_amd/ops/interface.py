"""Autograd interface over the gfx950 HIP kernels with a CPU reference path.

Dispatch policy:
- CUDA (= ROCm/HIP) tensors -> the _lumina_hip extension. If the extension is
  missing on a GPU machine this raises loudly (no silent eager fallback) unless
  LUMINA_ALLOW_FALLBACK=1 is set.
- CPU tensors -> the pure-PyTorch reference implementations (ops/reference.py).

Every Function here has a hand-written backward on both paths; numerics tests
(tests/test_ops_gpu.py) compare the HIP path against fp32 references.
"""

from __future__ import annotations

import math
import os
from typing import Optional

import torch

from . import reference as ref

_EXT = None
_TRIED = False


def _load_ext():
    global _EXT, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    from . import build_ext
    _EXT = build_ext.load_prebuilt()
    return _EXT


def get_ext():
    return _load_ext()


def has_ext() -> bool:
    return _load_ext() is not None


def _ext_or_raise():
    e = _load_ext()
    if e is None:
        if os.environ.get("LUMINA_ALLOW_FALLBACK") == "1":
            return None
        raise RuntimeError(
            "luminaai_amd: tensor is on GPU but the _lumina_hip extension is not "
            "built. Run `python -m luminaai_amd.ops.build_ext` (or __graft_entry__"
            ".build()). Set LUMINA_ALLOW_FALLBACK=1 to allow the slow eager path.")
    return e


def use_hip(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    return _ext_or_raise() is not None


# ------------------------------------------------------------------ RMSNorm
class RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        if use_hip(x):
            ext = get_ext()
            y, inv = ext.rmsnorm_fwd(x.contiguous(), weight.contiguous(), eps, True)
        else:
            y, inv = ref.rmsnorm_fwd_train(x, weight, eps)
        ctx.save_for_backward(x, weight, inv)
        return y

    @staticmethod
    def backward(ctx, gy):
        x, weight, inv = ctx.saved_tensors
        if use_hip(x):
            dx, dw = get_ext().rmsnorm_bwd(gy.contiguous(), x.contiguous(),
                                           weight.contiguous(), inv)
        else:
            dx, dw = ref.rmsnorm_bwd(gy, x, weight, inv)
        return dx, dw.to(weight.dtype), None


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    if torch.is_grad_enabled() and (x.requires_grad or weight.requires_grad):
        return RMSNormFn.apply(x, weight, eps)
    if use_hip(x):
        y, _ = get_ext().rmsnorm_fwd(x.contiguous(), weight.contiguous(), eps, False)
        return y
    return ref.rmsnorm_fwd(x, weight, eps)


# -------------------------------------------------------------------- RoPE
class RoPEFn(torch.autograd.Function):
    """Rotate q and k ([B, S, H, D] layout) in one launch. Backward is the
    inverse rotation (conj=True)."""

    @staticmethod
    def forward(ctx, q, k, cos, sin, pos, pos_offset):
        ctx.pos_offset = pos_offset
        if use_hip(q):
            oq, ok = get_ext().rope_fwd(q.contiguous(), k.contiguous(), cos, sin,
                                        pos, pos_offset, False)
            ctx.save_for_backward(cos, sin, pos if pos is not None else torch.empty(0))
            ctx.has_pos = pos is not None
            return oq, ok
        ctx.save_for_backward(cos, sin, pos if pos is not None else torch.empty(0))
        ctx.has_pos = pos is not None
        return _rope_ref_bshd(q, k, cos, sin, pos, pos_offset, conj=False)

    @staticmethod
    def backward(ctx, gq, gk):
        cos, sin, pos = ctx.saved_tensors
        pos = pos if ctx.has_pos else None
        if use_hip(gq):
            dq, dk = get_ext().rope_fwd(gq.contiguous(), gk.contiguous(), cos, sin,
                                        pos, ctx.pos_offset, True)
        else:
            dq, dk = _rope_ref_bshd(gq, gk, cos, sin, pos, ctx.pos_offset, conj=True)
        return dq, dk, None, None, None, None


def _rope_ref_bshd(q, k, cos, sin, pos, pos_offset, conj):
    """Reference rotation on [B, S, H, D] tensors (fp32 math)."""
    B, S, _, D = q.shape
    half = D // 2
    if pos is not None:
        c = cos[pos.view(B, S).long()]  # [B, S, half]
        s = sin[pos.view(B, S).long()]
    else:
        c = cos[pos_offset:pos_offset + S].unsqueeze(0).expand(B, S, half)
        s = sin[pos_offset:pos_offset + S].unsqueeze(0).expand(B, S, half)
    if conj:
        s = -s
    c = c.unsqueeze(2)  # [B, S, 1, half]
    s = s.unsqueeze(2)

    D = q.shape[-1]
    half = D // 2

    def rot(x):
        x32 = x.float()
        x1 = x32[..., :half]
        x2 = x32[..., half:2 * half]
        parts = [x1 * c - x2 * s, x2 * c + x1 * s]
        if D % 2:  # odd head_dim: last element passes through
            parts.append(x32[..., 2 * half:])
        return torch.cat(parts, dim=-1).to(x.dtype)

    return rot(q), rot(k)


def rope(q, k, cos, sin, pos: Optional[torch.Tensor] = None, pos_offset: int = 0):
    return RoPEFn.apply(q, k, cos, sin, pos, pos_offset)


# ------------------------------------------------------------------ SwiGLU
class SwiGLUFn(torch.autograd.Function):
    """y = silu(gate) * up on 2-D views (gate/up may be the two halves of a
    fused gate_up projection, sharing storage with row stride 2I)."""

    @staticmethod
    def forward(ctx, gate, up):
        ctx.save_for_backward(gate, up)
        if use_hip(gate):
            return get_ext().swiglu_fwd(gate, up)
        return ref.swiglu_fwd(gate, up)

    @staticmethod
    def backward(ctx, gy):
        gate, up = ctx.saved_tensors
        if use_hip(gate):
            dg, du = get_ext().swiglu_bwd(gy.contiguous(), gate, up)
        else:
            dg, du = ref.swiglu_bwd(gy, gate, up)
        return dg, du


class SwiGLUFusedFn(torch.autograd.Function):
    """y = silu(gu[:, :I]) * gu[:, I:] on the FUSED [M, 2I] projection
    output.  Compared with swiglu() on two narrow() views, autograd never
    sees the slices: backward writes one [M, 2I] grad buffer directly
    instead of narrow-backward's zero-fill + slice copies + add (measured
    ~6% of the b1 training step in eager glue kernels)."""

    @staticmethod
    def forward(ctx, gu2):
        I = gu2.shape[1] // 2
        ctx.save_for_backward(gu2)
        return get_ext().swiglu_fwd(gu2.narrow(1, 0, I), gu2.narrow(1, I, I))

    @staticmethod
    def backward(ctx, gy):
        (gu2,) = ctx.saved_tensors
        return get_ext().swiglu_bwd_fused(gy.contiguous(), gu2)


def swiglu_fused(gu2: torch.Tensor) -> torch.Tensor:
    """SwiGLU over the fused gate_up output [..., 2I] -> [..., I]."""
    I = gu2.shape[-1] // 2
    flat = gu2.reshape(-1, 2 * I)
    if use_hip(flat) and flat.is_contiguous():
        y = SwiGLUFusedFn.apply(flat)
    else:
        y = SwiGLUFn.apply(flat.narrow(1, 0, I), flat.narrow(1, I, I))
    return y.view(*gu2.shape[:-1], I)


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    shape = gate.shape
    g2 = gate.reshape(-1, shape[-1]) if gate.dim() != 2 else gate
    u2 = up.reshape(-1, shape[-1]) if up.dim() != 2 else up
    y = SwiGLUFn.apply(g2, u2)
    return y.view(shape)


# ------------------------------------------------- fused CE + accuracy loss
class FusedCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels_i32, weights, ignore_index):
        ext = get_ext()
        lse, stats = ext.ce_fwd(logits, labels_i32, weights, ignore_index)
        ctx.save_for_backward(logits, labels_i32,
                              weights if weights is not None else torch.empty(0),
                              lse, stats)
        ctx.has_w = weights is not None
        ctx.ignore_index = ignore_index
        loss = stats[0] / stats[1].clamp_min(1e-8)
        ctx.mark_non_differentiable(stats)
        return loss, stats

    @staticmethod
    def backward(ctx, gloss, _gstats):
        logits, labels, weights, lse, stats = ctx.saved_tensors
        weights = weights if ctx.has_w else None
        dlogits = get_ext().ce_bwd(logits, labels, weights, lse, stats,
                                   gloss.reshape(1).float().contiguous(),
                                   ctx.ignore_index)
        return dlogits, None, None, None


def fused_cross_entropy(logits: torch.Tensor, labels: torch.Tensor,
                        loss_weights: Optional[torch.Tensor] = None,
                        ignore_index: int = -100):
    """Returns (loss, accuracy, n_valid). GPU: single-pass HIP kernel;
    CPU: reference. `labels` may be int64; cast happens here."""
    logits2 = logits.reshape(-1, logits.shape[-1])
    labels1 = labels.reshape(-1)
    w = loss_weights.reshape(-1).float() if loss_weights is not None else None
    if use_hip(logits2):
        loss, stats = FusedCEFn.apply(logits2.contiguous(),
                                      labels1.to(torch.int32).contiguous(),
                                      w.contiguous() if w is not None else None,
                                      ignore_index)
        acc = stats[2] / stats[3].clamp_min(1.0)
        return loss, acc, stats[3]
    return ref.fused_cross_entropy(logits2, labels1, w, ignore_index)


# ------------------------------------------------------- batched expert GEMM
class BatchedLinearFn(torch.autograd.Function):
    """bmm with an MI355X-native backward.

    torch's built-in BmmBackward computes grad_x = grad @ w.transpose(1, 2)
    as a strided batched GEMM with a transposed-B operand — that pattern
    memory-faults in this ROCm hipBLASLt/rocBLAS build for large bf16 batches
    (verified on MI355X: any K/N, batched-only, transposed-B only). grad_x
    therefore runs on the hand-written MFMA grouped NT-GEMM kernel
    (csrc/grouped_gemm.hip) — both operands row-major, K-contiguous
    fragments, no transpose copy. grad_w uses the transposed-A hipBLASLt
    form, which is fine.
    """

    @staticmethod
    def forward(ctx, x, w):
        ctx.save_for_backward(x, w)
        import os as _os
        # A/B knob only: the NN-form 256^2 kernel measured 36.3k vs
        # hipBLASLt's 39.2k tok/s on the b1 step (the tr16-read B path
        # loses to the library's NN form at these shapes) -- default off
        if _os.environ.get("LUMINA_GG8P_FWD") and use_hip(x) \
                and x.dtype == torch.bfloat16 and w.shape[2] % 8 == 0 \
                and x.shape[2] == w.shape[1]:
            K = x.shape[2]
            xp = x if K % 32 == 0 else \
                torch.nn.functional.pad(x, (0, 32 - K % 32))
            return get_ext().gg8p_nn(xp.contiguous(), w.contiguous())
        return torch.bmm(x, w)

    @staticmethod
    def backward(ctx, go):
        x, w = ctx.saved_tensors
        go = go.contiguous()
        gx = gw = None
        if ctx.needs_input_grad[0]:
            gx = grouped_gemm_nt(go, w)
        if ctx.needs_input_grad[1]:
            gw = torch.bmm(x.transpose(1, 2), go)
        return gx, gw


def expert_bmm(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Grouped expert GEMM [E, C, K] x [E, K, N] (hipBLASLt strided-batched
    forward; hand-written MFMA NT kernel for grad_x)."""
    return BatchedLinearFn.apply(x, w)


# ------------------------------------------------------- flash attention
class _FlashAttnFn(torch.autograd.Function):
    """Causal GQA flash attention on the hand-written CDNA4 kernels
    (csrc/attention.hip). Inputs are PACKED [B, H, S, DP] bf16 contiguous
    with DP = padded head dim (zero pad channels); GQA handled natively
    (no KV repeat). Forward saves LSE2 (log2-sum-exp); backward runs the
    delta / dkdv / dq kernel pipeline.

    Replaces torch SDPA -> AOTriton on the training path (reference:
    flash_attention_dao_cuda.py:1, scaled_masked_softmax_cuda.cu:1)."""

    @staticmethod
    def forward(ctx, q, k, v, scale):
        o, lse2 = get_ext().attn_fwd(q, k, v, scale)
        ctx.save_for_backward(q, k, v, o, lse2)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse2 = ctx.saved_tensors
        dq, dk, dv = get_ext().attn_bwd(q, k, v, o, do.contiguous(), lse2,
                                        ctx.scale)
        return dq, dk, dv, None


def _attn_pad_dim(d: int) -> Optional[int]:
    for dp in (64, 128, 160):
        if d <= dp:
            return dp
    return None


def can_flash_attention(q: torch.Tensor, dropout: float) -> bool:
    """True when the hand-written causal-GQA kernel path applies."""
    return (q.is_cuda and q.dtype == torch.bfloat16 and dropout == 0.0
            and _attn_pad_dim(q.shape[-1]) is not None and has_ext())


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    scale: Optional[float] = None) -> torch.Tensor:
    """Causal GQA attention. q [B,S,H,D], k/v [B,S,HKV,D] -> [B,S,H,D].

    Pads D up to the kernel's tile dim and repacks to [B,H,S,DP]; the pad
    and transpose run through autograd, so gradients flow back to the
    original layout automatically."""
    B, S, H, D = q.shape
    dp = _attn_pad_dim(D)
    if scale is None:
        scale = D ** -0.5
    pad = dp - D

    def pack(t):
        t = t.transpose(1, 2)
        if pad:
            t = torch.nn.functional.pad(t, (0, pad))
        return t.contiguous()

    o = _FlashAttnFn.apply(pack(q), pack(k), pack(v), scale)
    if pad:
        o = o[..., :D]
    return o.transpose(1, 2)


_PAD_CACHE: dict = {}


def invalidate_pad_cache():
    """Drop cached padded weights (call after in-place weight updates:
    custom HIP optimizer kernels don't bump torch version counters)."""
    _PAD_CACHE.clear()


def _pad_k_cached(w: torch.Tensor, pad: int) -> torch.Tensor:
    """Zero-pad the trailing (K) dim, cached per (id, _version): grad_x
    re-pads the SAME weight tensor every micro-batch (62 calls/step on b1
    -- each a fill + full copy of a ~150 MB tensor). The version counter
    invalidates after the optimizer's in-place update; a weakref guards
    id recycling (same pattern as ops/fp8.py's weight cache)."""
    import weakref
    key = id(w)
    ent = _PAD_CACHE.get(key)
    if ent is not None and ent[0]() is w and ent[1] == w._version:
        return ent[2]
    out = torch.nn.functional.pad(w, (0, pad))
    if len(_PAD_CACHE) > 256:
        _PAD_CACHE.clear()
    _PAD_CACHE[key] = (weakref.ref(w), w._version, out)
    return out


def grouped_gemm_nt(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """out[e] = a[e] @ b[e]^T with a [E,M,K], b [E,N,K] both row-major
    (contraction over the trailing dim). HIP MFMA kernel on GPU bf16;
    fp32-math fallback elsewhere.

    Ragged K (b1's h=1908) is zero-padded up to 64 so it rides the glds
    double-buffered fast path: the pad copies cost ~0.1 ms while the
    register-staged ragged kernel ran ~1.1 ms slower per call (round-2
    profile: 7.4% of the training step)."""
    if use_hip(a) and a.dtype == torch.bfloat16 and b.dtype == torch.bfloat16:
        K = a.shape[-1]
        if K % 64:
            pad = 64 - K % 64
            a = torch.nn.functional.pad(a, (0, pad))
            b = _pad_k_cached(b, pad)
        import os as _os
        if _os.environ.get("LUMINA_GG128"):     # A/B knob: 128^2 kernel
            return get_ext().grouped_gemm_nt(a.contiguous(), b.contiguous())
        # default: the 256^2 8-phase kernel -- at hipBLASLt parity in
        # standalone microbenches but +1.3% on the whole training step
        # (2x the FLOPs/byte of the 128^2 tile holds up better against
        # the step's L2/HBM contention; measured 39.2k vs 38.7k tok/s)
        return get_ext().gg8p_nt(a.contiguous(), b.contiguous())
    return torch.matmul(a, b.transpose(1, 2).contiguous())


# ----------------------------------------------------------- optimizer path
# --------------------------------------------------------------------------
# MoE dispatch / combine: capacity-bucketed token exchange where BOTH the
# forward and the backward of each op are pure row GATHERS (index_select).
# The generic-indexing path the first implementation used (index_put /
# index_add) made autograd emit `indexing_backward_kernel` — a sort+atomics
# scatter that measured 25% of step time on MI355X (profiles/). The inverse
# permutation is precomputed once per routing decision, so no atomics and no
# d2h syncs remain anywhere in the MoE hot path.
# --------------------------------------------------------------------------
class MoERoutingPlan:
    """Index plan for one routing decision (no autograd state).

    slot_tm [N*k]: token-major flat choice -> buffer slot (E*C == dropped)
    inv     [E*C]: buffer slot -> token-major flat choice (N*k == empty)
    src_tok [E*C]: buffer slot -> source token row (clamped; see fill_mask)
    fill_mask [E*C] bool: slot actually filled
    counts  [E]: tokens routed per expert (pre-capacity)
    """

    __slots__ = ("slot_tm", "inv", "src_tok", "fill_mask", "counts",
                 "num_experts", "capacity", "top_k")

    def __init__(self, slot_tm, inv, src_tok, fill_mask, counts, E, C, k):
        self.slot_tm = slot_tm
        self.inv = inv
        self.src_tok = src_tok
        self.fill_mask = fill_mask
        self.counts = counts
        self.num_experts = E
        self.capacity = C
        self.top_k = k


@torch.no_grad()
def moe_routing_plan(topi: torch.Tensor, num_experts: int,
                     capacity: int) -> MoERoutingPlan:
    """topi [N, k] long -> gather plan. Tokens beyond an expert's capacity
    are dropped (stable order: earlier tokens win), matching the reference's
    capacity_factor semantics (model.py:1219-1242)."""
    N, k = topi.shape
    Nk = N * k
    E, C = num_experts, capacity
    dev = topi.device
    flat_e = topi.reshape(-1)                        # token-major
    order = torch.argsort(flat_e, stable=True)
    sorted_e = flat_e[order]
    counts = torch.bincount(flat_e, minlength=E)
    offs = torch.cumsum(counts, 0) - counts
    pos = torch.arange(Nk, device=dev) - offs[sorted_e]
    dest_sorted = torch.where(pos < C, sorted_e * C + pos,
                              torch.full_like(pos, E * C))
    slot_tm = torch.empty(Nk, dtype=torch.long, device=dev)
    slot_tm[order] = dest_sorted
    inv = torch.full((E * C + 1,), Nk, dtype=torch.long, device=dev)
    inv[dest_sorted] = order                         # E*C catches drops
    inv = inv[:E * C]
    fill_mask = inv < Nk
    src_tok = torch.div(inv.clamp_max(Nk - 1), k, rounding_mode="floor")
    return MoERoutingPlan(slot_tm, inv, src_tok, fill_mask, counts, E, C, k)


class MoEDispatchFn(torch.autograd.Function):
    """buf[s] = xf[src_tok[s]] * fill[s]; backward gathers grad rows back
    per (token, choice) and reduces over k — no scatter. On GPU both
    directions run as single fused HIP kernels (csrc/moe.hip)."""

    @staticmethod
    def forward(ctx, xf, src_tok, fill_mask, slot_tm, k):
        if use_hip(xf):
            buf = get_ext().moe_gather_rows(xf.contiguous(), src_tok,
                                            fill_mask)
        else:
            buf = xf.index_select(0, src_tok) \
                * fill_mask.unsqueeze(1).to(xf.dtype)
        ctx.save_for_backward(slot_tm)
        ctx.k = k
        ctx.EC = buf.shape[0]
        return buf

    @staticmethod
    def backward(ctx, gbuf):
        (slot_tm,) = ctx.saved_tensors
        if use_hip(gbuf):
            return (get_ext().moe_dispatch_bwd(gbuf.contiguous(), slot_tm,
                                               ctx.k),
                    None, None, None, None)
        keep = (slot_tm < ctx.EC).unsqueeze(1).to(gbuf.dtype)
        g = gbuf.index_select(0, slot_tm.clamp_max(ctx.EC - 1)) * keep
        N = slot_tm.numel() // ctx.k
        return g.view(N, ctx.k, -1).sum(1), None, None, None, None


class MoECombineFn(torch.autograd.Function):
    """out[t] = sum_j w[t,j] * y[slot_tm[t,j]] (gather + k-reduce); backward
    for y is a gather via the inverse permutation."""

    @staticmethod
    def forward(ctx, y, w_tm, slot_tm, inv, src_tok, fill_mask, k):
        EC = y.shape[0]
        N = slot_tm.numel() // k
        if use_hip(y):
            out = get_ext().moe_combine_fwd(y.contiguous(),
                                            w_tm.float().contiguous(),
                                            slot_tm, k)
        else:
            keep = slot_tm < EC
            slot_c = slot_tm.clamp_max(EC - 1)
            wk = (w_tm * keep.to(w_tm.dtype)).to(y.dtype)
            y_sel = y.index_select(0, slot_c)
            out = (y_sel * wk.unsqueeze(1)).view(N, k, -1).sum(1)
        ctx.save_for_backward(y, w_tm, slot_tm, inv, src_tok, fill_mask)
        ctx.k = k
        return out

    @staticmethod
    def backward(ctx, gout):
        y, w_tm, slot_tm, inv, src_tok, fill_mask, = ctx.saved_tensors
        EC = y.shape[0]
        k = ctx.k
        Nk = slot_tm.numel()
        if use_hip(y):
            gout = gout.contiguous()
            w32 = w_tm.float().contiguous()
            grad_y = get_ext().moe_combine_bwd_y(gout, w32, inv, src_tok,
                                                 fill_mask)
            gw = get_ext().moe_combine_bwd_w(gout, y.contiguous(), slot_tm, k)
            return (grad_y, gw.to(w_tm.dtype), None, None, None, None, None)
        # grad_y[s] = gout[src_tok[s]] * w_tm[inv[s]] * fill[s]
        w_slot = w_tm[inv.clamp_max(Nk - 1)] * fill_mask.to(w_tm.dtype)
        grad_y = gout.index_select(0, src_tok) * w_slot.unsqueeze(1).to(gout.dtype)
        # grad_w[t,j] = <gout[t], y[slot_tm[t,j]]> * keep
        keep = slot_tm < EC
        y_sel = y.index_select(0, slot_tm.clamp_max(EC - 1))
        N = Nk // k
        gw = (y_sel.view(N, k, -1).float()
              * gout.unsqueeze(1).float()).sum(-1).reshape(-1)
        gw = gw * keep.to(gw.dtype)
        return grad_y.to(y.dtype), gw.to(w_tm.dtype), None, None, None, None, None


def moe_dispatch(xf: torch.Tensor, plan: MoERoutingPlan) -> torch.Tensor:
    """[N, h] -> capacity buffer [E*C, h]."""
    return MoEDispatchFn.apply(xf, plan.src_tok, plan.fill_mask,
                               plan.slot_tm, plan.top_k)


def moe_combine(y: torch.Tensor, w_tm: torch.Tensor,
                plan: MoERoutingPlan) -> torch.Tensor:
    """Weighted gather back to token order: [E*C, h] -> [N, h].
    w_tm: token-major routing weights [N*k] (fp32)."""
    return MoECombineFn.apply(y, w_tm, plan.slot_tm, plan.inv,
                              plan.src_tok, plan.fill_mask, plan.top_k)


def l2norm_sq(flat: torch.Tensor) -> torch.Tensor:
    if use_hip(flat):
        return get_ext().l2norm_sq(flat)
    return flat.float().pow(2).sum().reshape(1)


def adamw_step(master: torch.Tensor, grad: torch.Tensor, m: torch.Tensor,
               v: torch.Tensor, w_out: Optional[torch.Tensor], lr: float,
               beta1: float, beta2: float, eps: float, wd: float, step: int,
               gnorm_sq: Optional[torch.Tensor], max_norm: float,
               grad_scale: float = 1.0):
    """Fused clip+AdamW on flat buffers. CPU path mirrors the kernel math."""
    if use_hip(master):
        get_ext().adamw_step(master, grad, m, v, w_out, lr, beta1, beta2, eps,
                             wd, step, gnorm_sq, max_norm, grad_scale)
        return
    g = grad.float() * grad_scale
    if gnorm_sq is not None:
        gn = gnorm_sq.sum().sqrt().item() * grad_scale
        if not math.isfinite(gn):
            return  # NaN/Inf grads: skip the whole step (matches HIP kernel)
        if max_norm > 0 and gn > max_norm:
            g = g * (max_norm / (gn + 1e-6))
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bias1 = 1.0 / (1.0 - beta1 ** step)
    bias2 = 1.0 / (1.0 - beta2 ** step)
    denom = (v * bias2).sqrt().add_(eps)
    master.add_(-lr * ((m * bias1) / denom + wd * master))
    if w_out is not None:
        w_out.copy_(master.to(w_out.dtype))
