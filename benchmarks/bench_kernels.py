#!/usr/bin/env python3
"""Per-kernel microbenchmarks: HIP kernels vs PyTorch-ROCm eager.

Parity with the reference's benchmark harnesses
(/root/reference/Src/Main_Scripts/core/benchmark_transformer_ops.py,
training/benchmark_cuda_kernels.py) — CLI table + JSON, no plots.

Usage (GPU box):  python benchmarks/bench_kernels.py [--json out.json]
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from luminaai_amd import ops  # noqa: E402
from luminaai_amd.ops import reference as ref  # noqa: E402
from luminaai_amd.ops.interface import grouped_gemm_nt  # noqa: E402


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def run_all():
    dev = torch.device("cuda")
    dt = torch.bfloat16
    results = []

    def add(name, ms_hip, ms_ref, flops=None, bytes_=None):
        row = {"kernel": name, "hip_ms": round(ms_hip, 4),
               "eager_ms": round(ms_ref, 4),
               "speedup": round(ms_ref / ms_hip, 2)}
        if flops:
            row["hip_tflops"] = round(flops / ms_hip / 1e9, 1)
        if bytes_:
            row["hip_tbps"] = round(bytes_ / ms_hip / 1e9, 2)
        results.append(row)
        print(f"{name:28s} hip {ms_hip:8.3f} ms   eager {ms_ref:8.3f} ms   "
              f"x{row['speedup']:<6}"
              + (f" {row.get('hip_tflops', '')} TF" if flops else "")
              + (f" {row.get('hip_tbps', '')} TB/s" if bytes_ else ""))

    # RMSNorm: b1 shape
    N, H = 8 * 2048, 1908
    x = torch.randn(N, H, device=dev, dtype=dt, requires_grad=True)
    w = torch.randn(H, device=dev, dtype=dt)
    add("rmsnorm_fwd [16k,1908]",
        timeit(lambda: ops.rmsnorm(x.detach(), w, 1e-6)),
        timeit(lambda: ref.rmsnorm_fwd(x.detach(), w, 1e-6)),
        bytes_=2 * x.numel() * 2)

    # RoPE
    B, S, Hh, D = 8, 2048, 12, 159
    q = torch.randn(B, S, Hh, 160, device=dev, dtype=dt)
    k = torch.randn(B, S, 4, 160, device=dev, dtype=dt)
    cos, sin = ref.rope_cache(S, 160, device=dev)
    add("rope [8,2048,12+4,160]",
        timeit(lambda: ops.rope(q, k, cos, sin)),
        timeit(lambda: ref.rope_apply(q.transpose(1, 2), k.transpose(1, 2),
                                      cos, sin)),
        bytes_=2 * (q.numel() + k.numel()) * 2)

    # SwiGLU
    M, I = 16384, 5120
    g = torch.randn(M, I, device=dev, dtype=dt)
    u = torch.randn(M, I, device=dev, dtype=dt)
    add("swiglu_fwd [16k,5120]",
        timeit(lambda: ops.swiglu(g, u)),
        timeit(lambda: ref.swiglu_fwd(g, u)),
        bytes_=3 * M * I * 2)

    # fused CE
    V = 50304
    logits = torch.randn(4096, V, device=dev, dtype=dt)
    labels = torch.randint(0, V, (4096,), device=dev)
    add("fused_ce [4096,50304]",
        timeit(lambda: ops.fused_cross_entropy(logits, labels)),
        timeit(lambda: ref.fused_cross_entropy(logits, labels)),
        bytes_=logits.numel() * 2)

    # grouped NT GEMM: b1 expert backward shape, vs fp32-free workaround
    E, C, K2I, Hd = 8, 2560, 10240, 1908
    go = torch.randn(E, C, K2I, device=dev, dtype=dt)
    wt = torch.randn(E, Hd, K2I, device=dev, dtype=dt)
    flops = 2.0 * E * C * K2I * Hd
    add(f"grouped_nt [{E},{C},{Hd},{K2I}]",
        timeit(lambda: grouped_gemm_nt(go, wt), iters=20),
        timeit(lambda: torch.bmm(go, wt.transpose(1, 2).contiguous()),
               iters=20),
        flops=flops)

    # decode GEMV: b1 qkv shape vs hipBLASLt batch-1 linear
    Ng, Kg = 2544, 1908
    wg = torch.randn(Ng, Kg, device=dev, dtype=dt)
    xg = torch.randn(Kg, device=dev, dtype=dt)
    from luminaai_amd.ops import get_ext
    add(f"gemv [{Ng},{Kg}]",
        timeit(lambda: get_ext().gemv(xg, wg), iters=200),
        timeit(lambda: torch.nn.functional.linear(xg, wg), iters=200),
        bytes_=Ng * Kg * 2)

    # AdamW fused step (flat 1e8 params)
    n = 100_000_000
    master = torch.randn(n, device=dev, dtype=torch.float32)
    grad = torch.randn(n, device=dev, dtype=dt)
    m = torch.zeros(n, device=dev, dtype=torch.float32)
    v = torch.zeros(n, device=dev, dtype=torch.float32)
    wout = torch.empty(n, device=dev, dtype=dt)
    gn = torch.ones(1, device=dev)
    add("adamw_step [1e8 params]",
        timeit(lambda: ops.interface.adamw_step(
            master, grad, m, v, wout, 1e-4, 0.9, 0.95, 1e-8, 0.01, 10,
            gn, 1.0, 1.0), iters=10),
        timeit(lambda: (m.mul_(0.9).add_(grad.float(), alpha=0.1),
                        v.mul_(0.95).addcmul_(grad.float(), grad.float(),
                                              value=0.05)), iters=10),
        bytes_=n * 30)

    return results


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--json", default=None)
    args = ap.parse_args()
    assert torch.cuda.is_available(), "run on a GPU box"
    results = run_all()
    if args.json:
        with open(args.json, "w") as f:
            json.dump({"device": torch.cuda.get_device_name(0),
                       "results": results}, f, indent=2)


if __name__ == "__main__":
    main()
