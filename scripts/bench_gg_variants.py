import torch, time, sys
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from luminaai_amd.ops import get_ext
ext = get_ext()
torch.manual_seed(0)
E, M, N, K = 8, 2560, 1908, 10240
a = torch.randn(E, M, K, device="cuda", dtype=torch.bfloat16)
b = torch.randn(E, N, K, device="cuda", dtype=torch.bfloat16)
ref = torch.matmul(a.float(), b.float().transpose(1, 2))
def check(out, name):
    err = (out.float() - ref).abs().max() / ref.abs().max()
    print(name, "relerr", float(err))
def bench(fn, name, iters=20):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / iters
    tf = 2.0 * E * M * N * K / dt / 1e12
    print(f"{name}: {dt*1e3:.3f} ms  {tf:.0f} TF")
check(ext.grouped_gemm_nt(a, b), "v1")
check(ext.grouped_gemm_nt_v2(a, b, 0), "v2s0")
check(ext.grouped_gemm_nt_v2(a, b, 1), "v2s1")
bench(lambda: ext.grouped_gemm_nt(a, b), "v1-16x16-glds")
bench(lambda: ext.grouped_gemm_nt_v2(a, b, 0), "v2-32x32-swz0")
bench(lambda: ext.grouped_gemm_nt_v2(a, b, 1), "v2-32x32-swz1")
# ragged small
E2, M2, N2, K2 = 3, 200, 300, 128
a2 = torch.randn(E2, M2, K2, device="cuda", dtype=torch.bfloat16)
b2 = torch.randn(E2, N2, K2, device="cuda", dtype=torch.bfloat16)
r2 = torch.matmul(a2.float(), b2.float().transpose(1,2))
for mode in (0,1):
    o2 = ext.grouped_gemm_nt_v2(a2, b2, mode)
    print("ragged mode", mode, "relerr", float((o2.float()-r2).abs().max()/r2.abs().max()))
check(ext.grouped_gemm_nt_v3(a, b), "v3")
bench(lambda: ext.grouped_gemm_nt_v3(a, b), "v3-3buf-vmcnt")
check(ext.grouped_gemm_nt_v4(a, b), "v4")
bench(lambda: ext.grouped_gemm_nt_v4(a, b), "v4-2buf-raw")
# race screen: repeated runs must be deterministic
o1 = ext.grouped_gemm_nt_v4(a, b)
for _ in range(5):
    d = (ext.grouped_gemm_nt_v4(a, b) - o1).abs().max()
    assert float(d) == 0.0, f"nondeterminism {float(d)}"
print("v4 race-screen ok")
# unaligned-K (fallback path) shape: gx2 of the down projection
Eu, Cu, Nu, Ku = 8, 2560, 5120, 1908
au = torch.randn(Eu, Cu, Ku, device="cuda", dtype=torch.bfloat16)
bu = torch.randn(Eu, Nu, Ku, device="cuda", dtype=torch.bfloat16)
ru = torch.matmul(au.float(), bu.float().transpose(1, 2))
ou = ext.grouped_gemm_nt(au, bu)
print("unaligned relerr", float((ou.float()-ru).abs().max()/ru.abs().max()))
E, M, N, K = Eu, Cu, Nu, Ku
bench(lambda: ext.grouped_gemm_nt(au, bu), "nt-unalignedK", iters=20)
