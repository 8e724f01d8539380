"""Precision management for MI355X (gfx950).

Rebuild of the reference PrecisionManager (trainer.py:157-356, a 19-entry
registry). On CDNA4 the realities are simpler and better:
- bf16 is the training dtype (MFMA bf16 ~2.5 PF dense);
- fp16 is supported but has no advantage over bf16 on this chip;
- fp8 is OCP e4m3fn/e5m2 (NOT the MI300X fnuz variants) for GEMM paths;
- there is NO TF32/xf32 on gfx950 — fp32 matmul runs at the 157 TF f32 rate.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, Optional

import torch


@dataclass
class PrecisionSpec:
    name: str
    param_dtype: torch.dtype
    compute_dtype: torch.dtype
    needs_loss_scale: bool
    notes: str = ""


_REGISTRY: Dict[str, PrecisionSpec] = {
    "fp32": PrecisionSpec("fp32", torch.float32, torch.float32, False,
                          "exact f32; no xf32 fast path on gfx950"),
    "tf32": PrecisionSpec("tf32", torch.float32, torch.float32, False,
                          "alias of fp32 on gfx950 (no TF32 hardware)"),
    "bf16": PrecisionSpec("bf16", torch.bfloat16, torch.bfloat16, False,
                          "pure-bf16 weights + fp32 master in FlatAdamW"),
    "mixed_bf16": PrecisionSpec("mixed_bf16", torch.bfloat16, torch.bfloat16, False),
    "fp16": PrecisionSpec("fp16", torch.float16, torch.float16, True),
    "mixed_fp16": PrecisionSpec("mixed_fp16", torch.float16, torch.float16, True),
    "fp8": PrecisionSpec("fp8", torch.bfloat16, torch.bfloat16, False,
                         "bf16 weights; fp8 e4m3fn GEMM forward (ops/fp8.py)"),
    "fp8_e5m2": PrecisionSpec("fp8_e5m2", torch.bfloat16, torch.bfloat16, False,
                              "reserved: e5m2 variant of the fp8 path"),
    "fp64": PrecisionSpec("fp64", torch.float64, torch.float64, False,
                          "debugging only; no MFMA path"),
    # registry aliases kept for reference Config compatibility
    # (trainer.py:157-356 listed int8/int4/... — those were post-training
    # quantization targets, served here by the inference loaders, not a
    # training dtype): they map to bf16 training with a warning.
    "int8": PrecisionSpec("int8", torch.bfloat16, torch.bfloat16, False,
                          "alias->bf16 (PTQ is an inference concern)"),
    "int4": PrecisionSpec("int4", torch.bfloat16, torch.bfloat16, False,
                          "alias->bf16 (PTQ is an inference concern)"),
    "mixed": PrecisionSpec("mixed", torch.bfloat16, torch.bfloat16, False,
                           "alias of mixed_bf16"),
}


class PrecisionManager:
    def __init__(self, config):
        self.config = config
        name = config.precision
        if name == "auto":
            name = "bf16" if torch.cuda.is_available() else "fp32"
        if name not in _REGISTRY:
            raise ValueError(f"unknown precision '{name}'; have {list(_REGISTRY)}")
        self.spec = _REGISTRY[name]

    @property
    def param_dtype(self) -> torch.dtype:
        return self.spec.param_dtype

    @property
    def compute_dtype(self) -> torch.dtype:
        return self.spec.compute_dtype

    @property
    def needs_loss_scale(self) -> bool:
        return self.spec.needs_loss_scale

    def cast_model(self, model: torch.nn.Module) -> torch.nn.Module:
        return model.to(dtype=self.param_dtype)

    @staticmethod
    def available() -> Dict[str, PrecisionSpec]:
        return dict(_REGISTRY)
