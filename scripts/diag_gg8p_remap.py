"""gg8p 8192^3 regression diag: XCD remap on/off (env set by caller)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from luminaai_amd.ops import get_ext
e = get_ext()
for size in (4096, 8192):
    torch.manual_seed(0)
    A = (torch.rand(1, size, size, device="cuda", dtype=torch.bfloat16) * 2 - 1)
    B = torch.rand_like(A) * 2 - 1
    fl = 2.0 * size**3
    for _ in range(3): e.gg8p_nt(A, B)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): e.gg8p_nt(A, B)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 10
    print(f"{size}^3 remap={'off' if os.environ.get('LUMINA_GG8P_NOREMAP') else 'on'}: {fl/dt/1e12:.0f} TF")

# expert grad_x shapes
for (E, M, N, K) in ((8, 10240, 1908, 10240), (8, 10240, 5120, 1920)):
    torch.manual_seed(0)
    A = (torch.rand(E, M, (K + 63) // 64 * 64, device="cuda", dtype=torch.bfloat16) * 2 - 1)
    B = (torch.rand(E, N, A.shape[2], device="cuda", dtype=torch.bfloat16) * 2 - 1)
    fl = 2.0 * E * M * N * A.shape[2]
    for _ in range(3): e.gg8p_nt(A, B)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): e.gg8p_nt(A, B)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 10
    print(f"E{E} {M}x{N}x{A.shape[2]} remap={'off' if os.environ.get('LUMINA_GG8P_NOREMAP') else 'on'}: {fl/dt/1e12:.0f} TF")
