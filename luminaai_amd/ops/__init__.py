"""Fused op layer: hand-written CDNA4 (gfx950) HIP kernels with autograd,
plus pure-PyTorch CPU reference implementations."""

from . import reference
from .interface import (
    adamw_step,
    can_flash_attention,
    flash_attention,
    fused_cross_entropy,
    get_ext,
    has_ext,
    l2norm_sq,
    rmsnorm,
    rope,
    swiglu, swiglu_fused,
)
from .reference import rope_cache

__all__ = [
    "adamw_step", "can_flash_attention", "flash_attention",
    "fused_cross_entropy", "get_ext", "has_ext", "l2norm_sq",
    "reference", "rmsnorm", "rope", "rope_cache", "swiglu", "swiglu_fused",
]
