"""Localize flash-attention numerics errors: which rows/cols/tiles differ."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F

from luminaai_amd.ops import flash_attention

torch.manual_seed(1)
for (B, S, H, HKV, D) in [(1, 128, 1, 1, 159), (1, 128, 1, 1, 128),
                          (1, 128, 1, 1, 64), (2, 256, 6, 2, 159)]:
    q = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, S, HKV, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, S, HKV, D, device="cuda", dtype=torch.bfloat16)
    scale = D ** -0.5
    o = flash_attention(q, k, v, scale)
    rep = H // HKV
    ref = F.scaled_dot_product_attention(
        q.float().transpose(1, 2),
        k.float().transpose(1, 2).repeat_interleave(rep, 1),
        v.float().transpose(1, 2).repeat_interleave(rep, 1),
        is_causal=True, scale=scale).transpose(1, 2)
    err = (o.float() - ref).abs()
    print(f"shape {(B,S,H,HKV,D)}: max {err.max().item():.4f} "
          f"mean {err.mean().item():.6f}")
    if err.max().item() > 3e-2:
        # error by s-row (first head)
        e = err[0, :, 0, :]
        bad_rows = (e.max(dim=1).values > 3e-2).nonzero().flatten()
        bad_cols = (e.max(dim=0).values > 3e-2).nonzero().flatten()
        print("  bad rows:", bad_rows[:16].tolist(), "...", len(bad_rows))
        print("  bad cols:", bad_cols[:32].tolist(), "...", len(bad_cols))
        print("  col mod16 hist:",
              torch.bincount(bad_cols % 16, minlength=16).tolist())
        print("  row mod32 hist:",
              torch.bincount(bad_rows % 32, minlength=32).tolist())
