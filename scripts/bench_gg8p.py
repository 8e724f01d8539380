"""gg8p (256^2 pipelined) vs hipBLASLt and the round-1 128^2 kernel on the
real b1 expert-GEMM shapes."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from luminaai_amd.ops import get_ext
ext = get_ext()
torch.manual_seed(0)
dev, dt = "cuda", torch.bfloat16

def bench(fn, flops, name, iters=20):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); dtm = (time.perf_counter() - t0) / iters
    print(f"{name}: {dtm*1e3:.3f} ms  {flops/dtm/1e12:.0f} TF")

def check(out, ref, name, tol=3e-2):
    err = ((out.float() - ref).abs().max() / ref.abs().max()).item()
    status = "OK" if err < tol else "FAIL"
    print(f"{name} relerr {err:.4f} {status}")

E, C, h, I = 8, 2560, 1908, 10240
hp = 1920

# ---- NT: down fwd  x2[E,C,I] @ wdn'[E,h,I]^T -> [E,C,h]
x2 = torch.randn(E, C, I, device=dev, dtype=dt)
wd = torch.randn(E, h, I, device=dev, dtype=dt) * 0.05
ref = torch.matmul(x2.float(), wd.float().transpose(1, 2))
check(ext.gg8p_nt(x2, wd), ref, "nt-down-fwd")
bench(lambda: ext.gg8p_nt(x2, wd), 2.0*E*C*h*I, "gg8p  nt [EC,I]x[h,I]")
wdt = wd.transpose(1, 2).contiguous()
bench(lambda: torch.matmul(x2, wdt), 2.0*E*C*h*I, "blaslt nn-precopied")
bench(lambda: ext.grouped_gemm_nt(x2, wd), 2.0*E*C*h*I, "v1-128 nt")

# ---- NT: gate_up grad_x  go[E,C,2I] @ wgu[E,h,2I]^T -> [E,C,h]
go = torch.randn(E, C, 2*I, device=dev, dtype=dt)
wgu = torch.randn(E, h, 2*I, device=dev, dtype=dt) * 0.05
ref = torch.matmul(go.float(), wgu.float().transpose(1, 2))
check(ext.gg8p_nt(go, wgu), ref, "nt-gu-gradx")
bench(lambda: ext.gg8p_nt(go, wgu), 2.0*E*C*h*2*I, "gg8p  nt [EC,2I]x[h,2I]")
bench(lambda: ext.grouped_gemm_nt(go, wgu), 2.0*E*C*h*2*I, "v1-128 nt")

# ---- NN: gate_up fwd  xpad[E,C,1920] @ wgu_kn[E,1908,2I] -> [E,C,2I]
xp = torch.zeros(E, C, hp, device=dev, dtype=dt)
xp[..., :h] = torch.randn(E, C, h, device=dev, dtype=dt)
wgu_kn = torch.randn(E, h, 2*I, device=dev, dtype=dt) * 0.05
ref = torch.matmul(xp[..., :h].float(), wgu_kn.float())
check(ext.gg8p_nn(xp, wgu_kn), ref, "nn-gu-fwd")
bench(lambda: ext.gg8p_nn(xp, wgu_kn), 2.0*E*C*h*2*I, "gg8p  nn [EC,1920]x[1908,2I]")
bench(lambda: torch.matmul(xp[..., :h], wgu_kn), 2.0*E*C*h*2*I, "blaslt nn")

# ---- NN: down grad_x  gopad[E,C,1920] @ wd'[E,1908,I] -> [E,C,I]
gop = torch.zeros(E, C, hp, device=dev, dtype=dt)
gop[..., :h] = torch.randn(E, C, h, device=dev, dtype=dt)
ref = torch.matmul(gop[..., :h].float(), wd.float())
check(ext.gg8p_nn(gop, wd), ref, "nn-down-gradx")
bench(lambda: ext.gg8p_nn(gop, wd), 2.0*E*C*h*I, "gg8p  nn [EC,1920]x[1908,I]")

# ---- square reference shapes (guide template: ~1330 TF @4k random)
for NK in (4096, 8192):
    a = torch.randn(1, NK, NK, device=dev, dtype=dt)
    b = torch.randn(1, NK, NK, device=dev, dtype=dt)
    bench(lambda: ext.gg8p_nt(a, b), 2.0*NK**3, f"gg8p nt {NK}^3")
    bench(lambda: ext.grouped_gemm_nt(a, b), 2.0*NK**3, f"v1   nt {NK}^3")
    del a, b
