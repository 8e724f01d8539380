"""gg8t (8-phase template) vs gg8p vs hipBLASLt: refcheck + rates."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from luminaai_amd.ops import get_ext
e = get_ext()

def rate(fn, flops, iters=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    return flops / dt / 1e12

def check(E, M, N, K, label):
    torch.manual_seed(0)
    A = (torch.rand(E, M, K, device="cuda", dtype=torch.bfloat16) * 2 - 1)
    B = (torch.rand(E, N, K, device="cuda", dtype=torch.bfloat16) * 2 - 1)
    ref = torch.matmul(A.float(), B.float().transpose(1, 2))
    for pf2 in (1, 2, 4):
        O = e.gg8t_nt(A, B, pf2)
        err = (O.float() - ref).abs().max().item()
        den = ref.abs().max().item()
        print(f"{label} pf2={pf2}: rel-max err {err/den:.2e}",
              "OK" if err/den < 2e-2 else "FAIL")
    fl = 2.0 * E * M * N * K
    r2 = rate(lambda: e.gg8t_nt(A, B, 2), fl)
    rp = rate(lambda: e.gg8p_nt(A, B), fl)
    print(f"{label}: gg8t(barrier) {r2:.0f}  gg8p {rp:.0f} TF")

check(1, 512, 512, 256, "small")
check(1, 4096, 4096, 4096, "4096^3")
check(8, 2560, 1908, 10240, "gate_up-gx/4")
check(8, 2560, 5120, 1920, "down-gx/4")
