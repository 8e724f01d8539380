"""Microbench the real expert/projection GEMM shapes on hipBLASLt to find
which patterns underperform (candidates for hand-written MFMA kernels)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def t(fn, iters=20, warmup=5):
    for _ in range(warmup): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

E, C, h, I2, I = 8, 2560, 1908, 10240, 5120  # b1 mb8-ish expert shapes
dt = torch.bfloat16
x = torch.randn(E, C, h, device="cuda", dtype=dt)
w = torch.randn(E, h, I2, device="cuda", dtype=dt)
go = torch.randn(E, C, I2, device="cuda", dtype=dt)
act = torch.randn(E, C, I, device="cuda", dtype=dt)
wd = torch.randn(E, I, h, device="cuda", dtype=dt)

fl = 2.0 * E * C * h * I2
d = t(lambda: torch.bmm(x, w));                 print(f"NN fwd gate_up: {d*1e3:7.3f} ms {fl/d/1e12:6.0f} TF")
d = t(lambda: torch.bmm(x.transpose(1,2), go)); print(f"TN gw gate_up : {d*1e3:7.3f} ms {fl/d/1e12:6.0f} TF")
fl2 = 2.0 * E * C * I * h
d = t(lambda: torch.bmm(act, wd));              print(f"NN fwd down   : {d*1e3:7.3f} ms {fl2/d/1e12:6.0f} TF")
gd = torch.randn(E, C, h, device="cuda", dtype=dt)
d = t(lambda: torch.bmm(act.transpose(1,2), gd)); print(f"TN gw down    : {d*1e3:7.3f} ms {fl2/d/1e12:6.0f} TF")

# dense projection shapes (token-major big GEMMs), mb16: 32768 tokens
T = 32768
xx = torch.randn(T, h, device="cuda", dtype=dt)
wq = torch.randn(2544, h, device="cuda", dtype=dt)
fl3 = 2.0 * T * h * 2544
d = t(lambda: torch.nn.functional.linear(xx, wq)); print(f"linear qkv fwd: {d*1e3:7.3f} ms {fl3/d/1e12:6.0f} TF")
g3 = torch.randn(T, 2544, device="cuda", dtype=dt)
d = t(lambda: g3.t() @ xx);                     print(f"linear qkv gw : {d*1e3:7.3f} ms {fl3/d/1e12:6.0f} TF")
d = t(lambda: g3 @ wq);                         print(f"linear qkv gx : {d*1e3:7.3f} ms {fl3/d/1e12:6.0f} TF")
wl = torch.randn(50304, h, device="cuda", dtype=dt)
fl4 = 2.0 * T * h * 50304
d = t(lambda: torch.nn.functional.linear(xx, wl), iters=5); print(f"lm_head fwd   : {d*1e3:7.3f} ms {fl4/d/1e12:6.0f} TF")
