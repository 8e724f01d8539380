"""Environment probing and validation, ROCm-first.

Rebuild of /root/reference/Src/Main_Scripts/utils/environment.py:11-563
(get_system_info :11, validate_environment :145, estimate_training_time :245,
device pick :392) with ROCm/MI355X probing instead of CUDA
compute-capability checks."""

from __future__ import annotations

import os
import platform
import shutil
import sys
from typing import Dict, List, Optional

import torch

try:
    import psutil
    _HAS_PSUTIL = True
except ImportError:
    _HAS_PSUTIL = False

# MI355X headline figures used for time estimation (dense bf16; fp8 doubles)
MI355X_BF16_TFLOPS = 2500.0
DEFAULT_MFU = 0.35


def get_system_info() -> Dict:
    info = {
        "platform": platform.platform(),
        "python": sys.version.split()[0],
        "torch": torch.__version__,
        "rocm_hip_version": getattr(torch.version, "hip", None),
        "cuda_available": torch.cuda.is_available(),
        "device_count": torch.cuda.device_count() if torch.cuda.is_available() else 0,
        "cpu_count": os.cpu_count(),
    }
    if _HAS_PSUTIL:
        vm = psutil.virtual_memory()
        info["ram_gb"] = round(vm.total / 1e9, 1)
        info["ram_available_gb"] = round(vm.available / 1e9, 1)
    if torch.cuda.is_available():
        p = torch.cuda.get_device_properties(0)
        info["gpu_name"] = p.name
        info["gpu_arch"] = getattr(p, "gcnArchName", "unknown")
        info["gpu_memory_gb"] = round(p.total_memory / 1e9, 1)
        info["gpu_multi_processor_count"] = p.multi_processor_count
    info["rocm_smi"] = shutil.which("rocm-smi") is not None
    info["hipcc"] = shutil.which("hipcc") is not None
    return info


def validate_environment(config=None) -> Dict:
    """Returns {"ok": bool, "errors": [...], "warnings": [...]}
    (reference environment.py:145-244)."""
    errors: List[str] = []
    warnings: List[str] = []
    info = get_system_info()

    if not info["cuda_available"]:
        warnings.append("no ROCm GPU visible — training will run on CPU")
    elif "gfx950" not in str(info.get("gpu_arch", "")):
        warnings.append(f"GPU arch {info.get('gpu_arch')} is not gfx950 "
                        "(MI355X); HIP kernels are compiled for gfx950")
    from ..ops import has_ext
    if info["cuda_available"] and not has_ext():
        errors.append("HIP extension _lumina_hip not built — run "
                      "python __graft_entry__.py (build)")
    if config is not None:
        if config.precision in ("fp8", "bf16", "fp16") and \
                not info["cuda_available"]:
            warnings.append(f"{config.precision} requested without a GPU; "
                            "tests fall back to fp32")
        need_gb = config.estimate_memory_gb()
        have_gb = info.get("gpu_memory_gb", 0)
        if info["cuda_available"] and need_gb > have_gb:
            errors.append(f"estimated {need_gb:.0f} GB needed but GPU has "
                          f"{have_gb:.0f} GB; lower batch or raise ZeRO stage")
    return {"ok": not errors, "errors": errors, "warnings": warnings,
            "info": info}


def estimate_training_time(config, dataset_tokens: int,
                           n_gpus: int = 1, mfu: float = DEFAULT_MFU) -> Dict:
    """Chinchilla-style wall-clock estimate from 6*N*D FLOPs
    (reference environment.py:245-391, hardware table replaced by the
    MI355X figure)."""
    n_active = config.estimate_active_params()
    total_flops = 6.0 * n_active * dataset_tokens * config.num_epochs
    flops_per_sec = MI355X_BF16_TFLOPS * 1e12 * mfu * n_gpus
    seconds = total_flops / flops_per_sec
    return {
        "active_params": n_active,
        "total_flops": total_flops,
        "assumed_mfu": mfu,
        "est_seconds": seconds,
        "est_hours": seconds / 3600.0,
        "est_days": seconds / 86400.0,
    }


def pick_device(prefer: Optional[str] = None) -> torch.device:
    if prefer:
        return torch.device(prefer)
    if torch.cuda.is_available():
        return torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0)))
    return torch.device("cpu")
