"""Weight-only int8 / int4 quantization for inference.

MI355X-native replacement for the reference QuantizationManager
(reference trainer.py:575-802: bnb Linear8bitLt swap :658-679, GPTQ 4-bit
config :681-710, quanto int8/int4 :712-736 — its bnb dispatch called a
method that was never defined and raised AttributeError). Here the
quantized formats are real and self-contained:

- int8: per-output-channel symmetric scales. 2x smaller weights than bf16.
- int4: group-wise scales (group_size along the input dim), two nibbles
  packed per byte. ~4x smaller weights.

Decode on MI355X is HBM-bandwidth-bound (profiles/r01_SUMMARY.md: the
batch-1 GEMV chain), so halving/quartering weight bytes directly raises
decode tokens/s ceilings. This module dequantizes at the GEMM input
(memory savings are realised at rest and over PCIe/xGMI transfers); the
fused dequant-GEMV HIP kernel is a round-2 item (ROADMAP.md).
Quantization is inference-only: modules register buffers, not Parameters.
"""

from __future__ import annotations

from typing import Tuple

import torch
import torch.nn as nn

INT8_MAX = 127.0
INT4_MAX = 7.0


# ---------------------------------------------------------------- int8
def quantize_int8(w: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-output-channel symmetric int8: w [N, K] -> (q int8 [N, K],
    scale fp32 [N]) with w ~= q * scale[:, None]."""
    amax = w.abs().amax(dim=1, keepdim=True).float().clamp_min(1e-12)
    scale = amax / INT8_MAX
    q = (w.float() / scale).round().clamp(-INT8_MAX, INT8_MAX).to(torch.int8)
    return q, scale.squeeze(1)


def dequantize_int8(q: torch.Tensor, scale: torch.Tensor,
                    dtype: torch.dtype = torch.bfloat16) -> torch.Tensor:
    return (q.float() * scale[:, None]).to(dtype)


class Int8Linear(nn.Module):
    """Inference Linear with int8 weight storage (per-channel scales)."""

    def __init__(self, in_features: int, out_features: int,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.compute_dtype = dtype
        self.register_buffer("weight_q",
                             torch.zeros(out_features, in_features,
                                         dtype=torch.int8))
        self.register_buffer("scale", torch.ones(out_features))

    @classmethod
    def from_linear(cls, lin: nn.Linear) -> "Int8Linear":
        assert lin.bias is None, "bias-free Linears only (model uses none)"
        mod = cls(lin.in_features, lin.out_features,
                  dtype=lin.weight.dtype if lin.weight.dtype.is_floating_point
                  else torch.bfloat16)
        q, s = quantize_int8(lin.weight.detach())
        mod.weight_q.copy_(q)
        mod.scale.copy_(s)
        return mod.to(lin.weight.device)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        w = dequantize_int8(self.weight_q, self.scale, self.compute_dtype)
        return torch.nn.functional.linear(x, w)

    def extra_repr(self) -> str:
        return f"in={self.in_features}, out={self.out_features}, wbits=8"


# ---------------------------------------------------------------- int4
def quantize_int4(w: torch.Tensor,
                  group_size: int = 128) -> Tuple[torch.Tensor, torch.Tensor]:
    """Group-wise symmetric int4: w [N, K] (K % group_size == 0) ->
    (packed uint8 [N, K/2], scale fp32 [N, K/group_size]). Values are
    stored biased by 8 (0..15), two per byte, low nibble first."""
    N, K = w.shape
    assert K % group_size == 0 and K % 2 == 0
    g = w.float().reshape(N, K // group_size, group_size)
    amax = g.abs().amax(dim=2, keepdim=True).clamp_min(1e-12)
    scale = amax / INT4_MAX
    q = (g / scale).round().clamp(-INT4_MAX, INT4_MAX).to(torch.int8)
    q = (q.reshape(N, K) + 8).to(torch.uint8)       # 1..15 biased
    packed = (q[:, 0::2] | (q[:, 1::2] << 4)).contiguous()
    return packed, scale.squeeze(2)


def dequantize_int4(packed: torch.Tensor, scale: torch.Tensor,
                    group_size: int = 128,
                    dtype: torch.dtype = torch.bfloat16) -> torch.Tensor:
    N = packed.shape[0]
    K = packed.shape[1] * 2
    q = torch.empty(N, K, dtype=torch.int8, device=packed.device)
    q[:, 0::2] = (packed & 0xF).to(torch.int8)
    q[:, 1::2] = (packed >> 4).to(torch.int8)
    q = q - 8
    w = q.float().reshape(N, K // group_size, group_size) * scale[:, :, None]
    return w.reshape(N, K).to(dtype)


class Int4Linear(nn.Module):
    """Inference Linear with packed int4 weights (group-wise scales)."""

    def __init__(self, in_features: int, out_features: int,
                 group_size: int = 128,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        assert in_features % group_size == 0
        self.in_features = in_features
        self.out_features = out_features
        self.group_size = group_size
        self.compute_dtype = dtype
        self.register_buffer("weight_q",
                             torch.zeros(out_features, in_features // 2,
                                         dtype=torch.uint8))
        self.register_buffer("scale",
                             torch.ones(out_features,
                                        in_features // group_size))

    @classmethod
    def from_linear(cls, lin: nn.Linear, group_size: int = 128) -> "Int4Linear":
        assert lin.bias is None
        mod = cls(lin.in_features, lin.out_features, group_size=group_size,
                  dtype=lin.weight.dtype if lin.weight.dtype.is_floating_point
                  else torch.bfloat16)
        q, s = quantize_int4(lin.weight.detach(), group_size)
        mod.weight_q.copy_(q)
        mod.scale.copy_(s)
        return mod.to(lin.weight.device)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        w = dequantize_int4(self.weight_q, self.scale, self.group_size,
                            self.compute_dtype)
        return torch.nn.functional.linear(x, w)

    def extra_repr(self) -> str:
        return (f"in={self.in_features}, out={self.out_features}, wbits=4, "
                f"group={self.group_size}")


# ---------------------------------------------------------------- MoE experts
def quantize_moe_experts(model: nn.Module) -> int:
    """Inference-only int8 storage for batched expert weights ([E, K, N],
    per-(expert, out-channel) scales over K). The experts hold most of a
    MoE model's parameters; this removes the bf16/fp32 copies entirely
    (the Parameters are deleted — the layer forward reads the int8
    buffers). Returns the number of MoE layers converted."""
    n = 0
    for layer in getattr(model, "get_moe_layers", lambda: [])():
        if getattr(layer, "use_int8_weights", False):
            continue
        for name in ("w_gate_up", "w_down"):
            w = getattr(layer, name).detach()            # [E, K, N]
            amax = w.abs().amax(dim=1, keepdim=True).float().clamp_min(1e-12)
            scale = amax / INT8_MAX                      # [E, 1, N]
            q = (w.float() / scale).round().clamp(
                -INT8_MAX, INT8_MAX).to(torch.int8)
            delattr(layer, name)                         # drop the Parameter
            layer.register_buffer(name + "_q", q)
            layer.register_buffer(name + "_scale", scale)
        layer.use_int8_weights = True
        n += 1
    return n


# ---------------------------------------------------------------- manager
def quantize_model(model: nn.Module, mode: str = "int8",
                   min_dim: int = 64, group_size: int = 128) -> int:
    """Swap eligible bias-free nn.Linear modules in place. mode: "int8" |
    "int4" | "fp8" (fp8 delegates to ops/fp8.py and stays trainable).
    Returns the number of modules converted."""
    if mode == "fp8":
        from .fp8 import convert_linears_to_fp8
        return convert_linears_to_fp8(model, min_dim=min_dim)
    if mode not in ("int8", "int4"):
        raise ValueError(f"unknown quantization mode {mode!r}")
    n = 0
    if mode == "int8":
        n += quantize_moe_experts(model)
    for mod in model.modules():
        for name, child in list(mod.named_children()):
            if type(child) is not nn.Linear or child.bias is not None:
                continue
            if min(child.weight.shape) < min_dim:
                continue
            if mode == "int4" and child.in_features % group_size != 0:
                continue
            new = (Int8Linear.from_linear(child) if mode == "int8"
                   else Int4Linear.from_linear(child, group_size))
            setattr(mod, name, new)
            n += 1
    return n


def quantized_model_bytes(model: nn.Module) -> int:
    """Weight bytes of the model as it stands (quantized buffers counted
    at their stored width) — the memory-footprint report the reference's
    QuantizationManager printed."""
    total = 0
    for p in model.parameters():
        total += p.numel() * p.element_size()
    for b in model.buffers():
        total += b.numel() * b.element_size()
    return total
