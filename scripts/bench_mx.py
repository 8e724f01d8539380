"""MX-fp8 grouped GEMM vs bf16 hipBLASLt on the b1 expert-forward shapes."""
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
    torch.cuda.synchronize(); d = (time.perf_counter() - t0) / iters
    print(f"{name}: {d*1e3:.3f} ms  {flops/d/1e12:.0f} TF")

E, C, h, I = 8, 2560, 1908, 10240
# gate_up fwd: x [E,C,1908] @ w [E,1908,20480]
x = torch.randn(E, C, h, device=dev, dtype=dt)
w = torch.randn(E, h, 2*I, device=dev, dtype=dt) * 0.05
flops = 2.0*E*C*h*2*I
qx, sx = ext.mx_quant_rows(x, 0)
from luminaai_amd.ops.fp8 import _mx_quantized_weight
wq, ws = _mx_quantized_weight(w)
ref = torch.matmul(x.float(), w.float())
out = ext.gg_mx_nt(qx, sx, wq, ws)
rel = ((out.float()-ref).abs().max()/ref.abs().max()).item()
print(f"gate_up relerr {rel:.4f} {'OK' if rel < 0.06 else 'FAIL'}")
bench(lambda: ext.gg_mx_nt(qx, sx, wq, ws), flops, "mx gemm only")
bench(lambda: ext.gg_mx_nt(ext.mx_quant_rows(x, 0)[0], sx, wq, ws), flops, "mx quant_x+gemm")
bench(lambda: torch.matmul(x, w), flops, "blaslt bf16 nn")
# quant cost alone
bench(lambda: ext.mx_quant_rows(x, 0), flops, "quant_rows (same-flops scale)")
bench(lambda: _mx_quantized_weight(torch.randn(1,256,256,device=dev,dtype=dt)), 1e9, "wquant tiny (cache miss)")
# down fwd: x2 [E,C,I] @ wdn [E,I,h]
x2 = torch.randn(E, C, I, device=dev, dtype=dt)
wd = torch.randn(E, I, h, device=dev, dtype=dt) * 0.05
qx2, sx2 = ext.mx_quant_rows(x2, 0)
wdq, wds = _mx_quantized_weight(wd)
ref = torch.matmul(x2.float(), wd.float())
out = ext.gg_mx_nt(qx2, sx2, wdq, wds)
rel = ((out.float()-ref).abs().max()/ref.abs().max()).item()
print(f"down relerr {rel:.4f} {'OK' if rel < 0.06 else 'FAIL'}")
bench(lambda: ext.gg_mx_nt(qx2, sx2, wdq, wds), 2.0*E*C*I*h, "mx down gemm")
bench(lambda: torch.matmul(x2, wd), 2.0*E*C*I*h, "blaslt down bf16")
