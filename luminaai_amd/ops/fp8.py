"""fp8 (OCP e4m3fn) GEMM path for gfx950.

The reference's fp8 support was registry-only and never reached a kernel
(reference trainer.py:157-356). Here fp8 is real: gfx950 MFMA runs OCP
e4m3fn at ~2x the bf16 rate (MI355X_MICROARCH.md §Matrix cores; NOT the
MI300X fnuz variants), reached through hipBLASLt via torch._scaled_mm.

Scheme: bf16 master weights; per-tensor dynamic scaling (amax / 448 for
e4m3fn) on both activations and weights; fp32 accumulate; bf16 output.
The backward runs in bf16 (standard "fp8 forward, bf16 backward" training
recipe — wgrad/dgrad keep full bf16 fidelity).
"""

from __future__ import annotations

import os
import weakref

import torch
import torch.nn as nn

E4M3_MAX = 448.0


def _amax_scale(t: torch.Tensor) -> torch.Tensor:
    amax = t.abs().amax().float().clamp_min(1e-12)
    return (E4M3_MAX / amax).clamp(max=1e12)


def quantize_e4m3(t: torch.Tensor):
    """Returns (fp8 tensor, inverse scale fp32 scalar tensor)."""
    s = _amax_scale(t)
    q = (t.float() * s).clamp(-E4M3_MAX, E4M3_MAX).to(torch.float8_e4m3fn)
    return q, (1.0 / s)


def _quantized_weight_2d(w: torch.Tensor):
    # id() keys can be recycled after a weight is freed -- verify the cache
    # entry still refers to THIS tensor object via a weakref before trusting
    # it (a stale hit returned a wrong-shape quantized weight).
    ent = _WCACHE.get(id(w))
    if ent is not None and ent[3] == _EPOCH and ent[0]() is w:
        return ent[1], ent[2]
    q, s = quantize_e4m3(w)
    _WCACHE[id(w)] = (weakref.ref(w), q, s, _EPOCH)
    return q, s


def _mx_quantized_weight_2d(w: torch.Tensor):
    """nn.Linear weight [N, K] (already the NT layout) -> rowwise MX
    quantization, cached per optimizer epoch."""
    ent = _WCACHE.get(("mx2", id(w)))
    if ent is not None and ent[3] == _EPOCH and ent[0]() is w:
        return ent[1], ent[2]
    from .interface import get_ext
    q, s = get_ext().mx_quant_rows(w.contiguous(), 0)
    _WCACHE[("mx2", id(w))] = (weakref.ref(w), q, s, _EPOCH)
    return q, s


class _FP8MatmulFn(torch.autograd.Function):
    """y = x @ w.T in fp8 forward; bf16 backward. Uses the hand-written
    MX-fp8 MFMA kernel (the only fp8 that beats bf16 on gfx950) when
    built; falls back to torch._scaled_mm per-tensor e4m3."""

    @staticmethod
    def forward(ctx, x, w):
        ctx.save_for_backward(x, w)
        shp = x.shape[:-1]
        x2 = x.reshape(-1, x.shape[-1])
        if mx_available() and x.dtype == torch.bfloat16:
            from .interface import get_ext
            ext = get_ext()
            qx, sx = ext.mx_quant_rows(x2.contiguous(), 0)
            wq, ws = _mx_quantized_weight_2d(w)
            y = ext.gg_mx_nt(qx.unsqueeze(0), sx, wq.unsqueeze(0), ws)[0]
            return y.reshape(*shp, w.shape[0])
        xq, xs = quantize_e4m3(x2)
        wq, ws = _quantized_weight_2d(w)
        y = torch._scaled_mm(xq, wq.t(), scale_a=xs, scale_b=ws,
                             out_dtype=torch.bfloat16)
        return y.reshape(*shp, w.shape[0])

    @staticmethod
    def backward(ctx, gy):
        x, w = ctx.saved_tensors
        gy2 = gy.reshape(-1, gy.shape[-1])
        gx = gw = None
        if ctx.needs_input_grad[0]:
            gx = (gy2 @ w.to(gy.dtype)).reshape_as(x)
        if ctx.needs_input_grad[1]:
            gw = gy2.t() @ x.reshape(-1, x.shape[-1]).to(gy.dtype)
        return gx, gw


class FP8Linear(nn.Linear):
    """Drop-in nn.Linear whose forward GEMM runs on fp8 MFMA.
    Falls back to bf16 matmul off-GPU or for tiny shapes."""

    fp8_min_dim = 64  # _scaled_mm needs %16 shapes; skip tiny layers

    def forward(self, x):
        w = self.weight
        if (x.is_cuda and self.bias is None
                and x.shape[-1] % 16 == 0 and w.shape[0] % 16 == 0
                and min(w.shape) >= self.fp8_min_dim):
            return _FP8MatmulFn.apply(x, w)
        return super().forward(x)


# quantized-weight cache: expert weights only change at optimizer steps, so
# the transpose+amax+cast of the (large) weight operand is done once per step
# (Trainer.optimizer_step calls invalidate_weight_cache); activations are
# quantized per call.
_WCACHE = {}
_EPOCH = 0


def invalidate_weight_cache():
    global _EPOCH
    _EPOCH += 1


def _quantized_weight(w: torch.Tensor):
    ent = _WCACHE.get(id(w))
    if ent is not None and ent[3] == _EPOCH and ent[0]() is w:
        return ent[1], ent[2]
    qs, ss = [], []
    for e in range(w.shape[0]):
        q, s = quantize_e4m3(w[e].t().contiguous())     # [N, K] row-major
        qs.append(q)
        ss.append(s)
    _WCACHE[id(w)] = (weakref.ref(w), qs, ss, _EPOCH)
    return qs, ss


class _FP8ExpertBmmFn(torch.autograd.Function):
    """Grouped expert GEMM with fp8 e4m3 forward (per-expert 2D scaled_mm —
    hipBLASLt has no batched fp8 entry; E launches amortise fine) and bf16
    backward through the MFMA NT kernel + transA bmm."""

    @staticmethod
    def forward(ctx, x, w):
        ctx.save_for_backward(x, w)
        E, C, K = x.shape
        N = w.shape[2]
        out = x.new_empty(E, C, N)
        wq, ws = _quantized_weight(w)
        for e in range(E):
            xq, xs = quantize_e4m3(x[e])
            out[e] = torch._scaled_mm(xq, wq[e].t(), scale_a=xs,
                                      scale_b=ws[e],
                                      out_dtype=torch.bfloat16)
        return out

    @staticmethod
    def backward(ctx, go):
        from .interface import grouped_gemm_nt
        x, w = ctx.saved_tensors
        go = go.contiguous()
        gx = gw = None
        if ctx.needs_input_grad[0]:
            gx = grouped_gemm_nt(go, w)
        if ctx.needs_input_grad[1]:
            gw = torch.bmm(x.transpose(1, 2), go)
        return gx, gw


def expert_bmm_fp8(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """fp8-forward grouped expert GEMM: x [E,C,K] @ w [E,K,N] -> [E,C,N].
    Uses the hand-written MX-fp8 MFMA kernel (2x bf16 rate) when built;
    falls back to per-expert torch._scaled_mm, then to bf16 bmm."""
    if x.is_cuda and x.dtype == torch.bfloat16 and mx_available() \
            and w.shape[2] % 8 == 0:
        return _MXExpertBmmFn.apply(x, w)
    if x.is_cuda and x.shape[-1] % 16 == 0 and w.shape[-1] % 16 == 0 \
            and x.shape[1] % 16 == 0:
        return _FP8ExpertBmmFn.apply(x, w)
    from .interface import expert_bmm
    return expert_bmm(x, w)


def convert_linears_to_fp8(model: nn.Module, min_dim: int = 64) -> int:
    """Swap eligible nn.Linear modules for FP8Linear in place
    (cf. the reference QuantizationManager's bnb Linear8bitLt swap,
    trainer.py:658-679). Returns the number converted."""
    n = 0
    for mod in model.modules():
        for name, child in list(mod.named_children()):
            if type(child) is nn.Linear and child.bias is None \
                    and min(child.weight.shape) >= min_dim:
                new = FP8Linear(child.in_features, child.out_features,
                                bias=False, device=child.weight.device,
                                dtype=child.weight.dtype)
                new.weight = child.weight
                setattr(mod, name, new)
                n += 1
    return n


# ---------------------------------------------------------------------------
# MX-fp8 rowwise path (hand-written gfx950 kernels, ops/csrc/mxfp8.hip):
# v_mfma_scale_f32_32x32x64_f8f6f4 runs at the ~5 PF fp8 rate -- the only
# fp8 that is actually faster than bf16 on this chip (plain e4m3 measured
# throughput-flat in round 1).  Forward-only; backward stays bf16.
def mx_available() -> bool:
    from .interface import get_ext, has_ext
    return has_ext() and hasattr(get_ext(), "gg_mx_nt")


def _mx_quantized_weight(w: torch.Tensor):
    """w [E, K, N] bf16 -> (qT [E, N, Kp] u8, s [E, N] u8), cached per
    optimizer epoch (weights only change at steps)."""
    ent = _WCACHE.get(("mx", id(w)))
    if ent is not None and ent[3] == _EPOCH and ent[0]() is w:
        return ent[1], ent[2]
    from .interface import get_ext
    qT, sc = get_ext().mx_quant_cols(w.contiguous(), 0)   # batched [E,K,N]
    _WCACHE[("mx", id(w))] = (weakref.ref(w), qT, sc, _EPOCH)
    return qT, sc


def _mx_quantized_weight_rows(w: torch.Tensor):
    """w [E, K, N] bf16 -> rowwise-quantized NATURAL layout (q [E, K, Np]
    u8, s [E, K] u8), cached per optimizer epoch.  This is the B operand
    of the MX dgrad GEMM: grad_x = go @ w^T contracts over N, and w's
    rows are already N-contiguous -- no transpose needed."""
    ent = _WCACHE.get(("mxr", id(w)))
    if ent is not None and ent[3] == _EPOCH and ent[0]() is w:
        return ent[1], ent[2]
    from .interface import get_ext
    q, sc = get_ext().mx_quant_rows(w.contiguous(), 0)
    _WCACHE[("mxr", id(w))] = (weakref.ref(w), q, sc, _EPOCH)
    return q, sc


# MX dgrad measured SLOWER end-to-end than bf16 dgrad (40.7k vs 41.5k
# tok/s on the b1 fp8 step): the per-call rowwise quantization of go
# costs more than the 1.3-1.5 PF GEMM saves.  bf16 dgrad is the default;
# LUMINA_FP8_MX_DGRAD=1 opts in (capability + memory-bound regimes).
_MX_DGRAD = bool(os.environ.get("LUMINA_FP8_MX_DGRAD"))


class _MXExpertBmmFn(torch.autograd.Function):
    """Grouped expert GEMM: MX-fp8 forward (+ optional MX dgrad, roadmap
    item 4a: grad_x contracts w's natural rows so the same NT kernel
    serves it), hipBLASLt bf16 for wgrad (needs a TN form the MX kernel
    doesn't have)."""

    @staticmethod
    def forward(ctx, x, w):
        ctx.save_for_backward(x, w)
        from .interface import get_ext
        ext = get_ext()
        E, C, K = x.shape
        qx, sx = ext.mx_quant_rows(x.contiguous(), 0)
        wq, ws = _mx_quantized_weight(w)
        return ext.gg_mx_nt(qx, sx, wq, ws)

    @staticmethod
    def backward(ctx, go):
        from .interface import get_ext, grouped_gemm_nt
        x, w = ctx.saved_tensors
        go = go.contiguous()
        gx = gw = None
        if ctx.needs_input_grad[0]:
            if _MX_DGRAD:
                ext = get_ext()
                qg, sg = ext.mx_quant_rows(go, 0)
                wq, ws = _mx_quantized_weight_rows(w)
                gx = ext.gg_mx_nt(qg, sg, wq, ws)
            else:
                gx = grouped_gemm_nt(go, w)
        if ctx.needs_input_grad[1]:
            gw = torch.bmm(x.transpose(1, 2), go)
        return gx, gw
