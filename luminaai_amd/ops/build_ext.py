"""In-tree build of the _lumina_hip extension for gfx950.

Kernels (*.hip) are compiled by hipcc with --offload-arch=gfx950 (pure HIP,
no torch headers, no hipify). The host bindings (bindings.cpp) are compiled
through torch.utils.cpp_extension and linked with the kernel objects.
Everything lands in luminaai_amd/ops/_build/ inside the repo so the .so
travels with the source tree to GPU boxes.
"""

from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

CSRC = Path(__file__).resolve().parent / "csrc"
BUILD_DIR = Path(__file__).resolve().parent / "_build"
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

KERNELS = ["rmsnorm.hip", "rope.hip", "swiglu.hip", "ce_loss.hip", "optim.hip",
           "grouped_gemm.hip", "moe.hip", "gemv.hip", "attention.hip",
           "gemm8p.hip",
    "gemm8t.hip", "mxfp8.hip", "decode.hip"]


def _newer(a: Path, b: Path) -> bool:
    return not b.exists() or a.stat().st_mtime > b.stat().st_mtime


def compile_kernels(verbose: bool = False):
    BUILD_DIR.mkdir(exist_ok=True)
    objs = []
    common = CSRC / "common.h"
    for k in KERNELS:
        src = CSRC / k
        if not src.exists():
            continue
        obj = BUILD_DIR / (k.replace(".hip", ".o"))
        if _newer(src, obj) or _newer(common, obj):
            cmd = [
                f"{ROCM}/bin/hipcc", f"--offload-arch={ARCH}", "-O3",
                "-std=c++17", "-fPIC", "-c", str(src), "-o", str(obj),
            ]
            if verbose:
                print(" ".join(cmd))
            subprocess.run(cmd, check=True)
        objs.append(str(obj))
    return objs


def build(verbose: bool = False):
    """Compile kernels + bindings; returns the loaded module."""
    from torch.utils import cpp_extension

    objs = compile_kernels(verbose=verbose)
    BUILD_DIR.mkdir(exist_ok=True)
    # ninja only tracks bindings.cpp; force a relink when any kernel .o is
    # newer than the linked extension.
    so = BUILD_DIR / "_lumina_hip.so"
    if so.exists() and any(_newer(Path(o), so) for o in objs):
        so.unlink()
    module = cpp_extension.load(
        name="_lumina_hip",
        sources=[str(CSRC / "bindings.cpp")],
        extra_cflags=["-O2", "-std=c++17"],
        # kernel .o files ride along as linker inputs
        extra_ldflags=objs + [f"-L{ROCM}/lib", "-lamdhip64"],
        extra_include_paths=[f"{ROCM}/include"],
        build_directory=str(BUILD_DIR),
        verbose=verbose,
        with_cuda=False,
    )
    return module


def load_prebuilt():
    """Import a previously built .so without invoking any compiler."""
    import importlib.util
    so = BUILD_DIR / "_lumina_hip.so"
    if not so.exists():
        return None
    spec = importlib.util.spec_from_file_location("_lumina_hip", str(so))
    mod = importlib.util.module_from_spec(spec)
    try:
        spec.loader.exec_module(mod)
    except ImportError:
        return None
    sys.modules["_lumina_hip"] = mod
    return mod


if __name__ == "__main__":
    build(verbose=True)
    print("built OK ->", BUILD_DIR / "_lumina_hip.so")
