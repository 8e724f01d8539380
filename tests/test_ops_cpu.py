"""CPU reference-op tests: semantics of every fused op vs plain torch autograd
(the same references the HIP kernels are tested against on GPU)."""

import math

import pytest
import torch
import torch.nn.functional as F

from luminaai_amd.ops import reference as ref
from luminaai_amd.ops import interface as K


def test_rmsnorm_fwd_matches_formula():
    torch.manual_seed(0)
    x = torch.randn(4, 33, dtype=torch.float64).float()
    w = torch.randn(33)
    y = ref.rmsnorm_fwd(x, w, 1e-6)
    expect = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-6) * w
    assert torch.allclose(y, expect, atol=1e-5)


def test_rmsnorm_bwd_matches_autograd():
    torch.manual_seed(1)
    x = torch.randn(8, 64, requires_grad=True)
    w = torch.randn(64, requires_grad=True)
    y = K.rmsnorm(x, w, 1e-6)
    g = torch.randn_like(y)
    y.backward(g)
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    y2 = x2 * torch.rsqrt(x2.pow(2).mean(-1, keepdim=True) + 1e-6) * w2
    y2.backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-4)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4)


def test_rope_rotation_preserves_norm():
    cos, sin = ref.rope_cache(16, 8)
    q = torch.randn(2, 16, 3, 8)
    k = torch.randn(2, 16, 2, 8)
    oq, ok = K.rope(q, k, cos, sin)
    assert torch.allclose(oq.norm(dim=-1), q.norm(dim=-1), atol=1e-4)
    assert oq.shape == q.shape and ok.shape == k.shape
    # position 0 is identity
    assert torch.allclose(oq[:, 0], q[:, 0], atol=1e-5)


def test_rope_backward_is_inverse_rotation():
    cos, sin = ref.rope_cache(8, 4)
    q = torch.randn(1, 8, 2, 4, requires_grad=True)
    k = torch.randn(1, 8, 2, 4, requires_grad=True)
    oq, ok = K.rope(q, k, cos, sin)
    (oq.sum() + ok.sum()).backward()
    # gradient of a rotation is the transposed (inverse) rotation of ones
    g = torch.ones_like(q)
    gq, gk = K.RoPEFn.apply(g, torch.ones_like(k), cos, sin, None, 0)
    # rotation by -theta of ones equals backward grad
    inv_q, _ = ref.rope_apply(g.transpose(1, 2), g.transpose(1, 2), cos, -sin)
    assert torch.allclose(q.grad, inv_q.transpose(1, 2), atol=1e-4)


def test_swiglu_matches_autograd():
    torch.manual_seed(2)
    g = torch.randn(16, 32, requires_grad=True)
    u = torch.randn(16, 32, requires_grad=True)
    y = K.swiglu(g, u)
    gy = torch.randn_like(y)
    y.backward(gy)
    g2 = g.detach().clone().requires_grad_(True)
    u2 = u.detach().clone().requires_grad_(True)
    (F.silu(g2) * u2).backward(gy)
    assert torch.allclose(g.grad, g2.grad, atol=1e-5)
    assert torch.allclose(u.grad, u2.grad, atol=1e-5)


def test_fused_ce_matches_torch():
    torch.manual_seed(3)
    logits = torch.randn(10, 50)
    labels = torch.randint(0, 50, (10,))
    labels[3] = -100
    loss, acc, n = ref.fused_cross_entropy(logits, labels)
    expect = F.cross_entropy(logits, labels, ignore_index=-100)
    assert torch.allclose(loss, expect, atol=1e-5)
    assert n == 9


def test_fused_ce_weighted():
    torch.manual_seed(4)
    logits = torch.randn(6, 20)
    labels = torch.randint(0, 20, (6,))
    w = torch.tensor([2.0, 1.0, 0.0, 1.0, 1.0, 3.0])
    loss, acc, n = ref.fused_cross_entropy(logits, labels, w)
    nll = F.cross_entropy(logits, labels, reduction="none")
    expect = (nll * w).sum() / w.sum()
    assert torch.allclose(loss, expect, atol=1e-5)


def test_topk_gating_normalized():
    logits = torch.randn(32, 8)
    wts, idx, probs = ref.topk_gating(logits, 2)
    assert torch.allclose(wts.sum(-1), torch.ones(32), atol=1e-5)
    assert (idx >= 0).all() and (idx < 8).all()
    assert torch.allclose(probs.sum(-1), torch.ones(32), atol=1e-5)


def test_load_balancing_loss_uniform_is_one():
    # perfectly uniform routing -> loss == 1.0 (E * E * (1/E) * (1/E))
    probs = torch.full((64, 4), 0.25)
    topi = torch.arange(64).remainder(4).unsqueeze(1)
    loss = ref.load_balancing_loss(probs, topi, 4)
    assert abs(float(loss) - 1.0) < 1e-5


def test_adamw_step_cpu_matches_torch_adamw():
    torch.manual_seed(5)
    p0 = torch.randn(100)
    g0 = torch.randn(100)
    # our flat implementation
    master = p0.clone()
    m = torch.zeros(100)
    v = torch.zeros(100)
    K.adamw_step(master, g0.clone(), m, v, None, lr=1e-2, beta1=0.9,
                 beta2=0.95, eps=1e-8, wd=0.0, step=1, gnorm_sq=None,
                 max_norm=0.0)
    # torch reference
    p = p0.clone().requires_grad_(True)
    opt = torch.optim.AdamW([p], lr=1e-2, betas=(0.9, 0.95), eps=1e-8,
                            weight_decay=0.0)
    p.grad = g0.clone()
    opt.step()
    assert torch.allclose(master, p.detach(), atol=1e-6)


def test_adamw_nan_skip():
    master = torch.ones(10)
    g = torch.full((10,), float("nan"))
    m = torch.zeros(10)
    v = torch.zeros(10)
    ns = torch.tensor([float("nan")])
    K.adamw_step(master, g, m, v, None, 1e-2, 0.9, 0.95, 1e-8, 0.0, 1, ns, 1.0)
    assert torch.allclose(master, torch.ones(10))  # step skipped


def test_adamw_clip():
    master = torch.zeros(4)
    g = torch.full((4,), 10.0)
    m = torch.zeros(4)
    v = torch.zeros(4)
    ns = (g.pow(2).sum()).reshape(1)
    K.adamw_step(master, g, m, v, None, lr=1.0, beta1=0.0, beta2=0.0,
                 eps=1e-8, wd=0.0, step=1, gnorm_sq=ns, max_norm=1.0)
    # after clip, g_i = 10 * (1/20) = 0.5 each; m = g; v = g^2; update = -lr*m/sqrt(v) ~ -1
    assert torch.allclose(master, torch.full((4,), -1.0), atol=1e-3)


def test_flat_adamw_offload_matches_resident():
    """offload=True (host-resident master/m/v) must produce identical
    weights to the resident optimizer. On CPU the staging copies are
    no-ops, but the state-placement and step plumbing are the same code
    that runs under ROCm."""
    import torch.nn as nn
    from luminaai_amd.training.optimizer import FlatAdamW
    torch.manual_seed(0)

    def build():
        torch.manual_seed(7)
        m = nn.Sequential(nn.Linear(16, 32), nn.SiLU(), nn.Linear(32, 16))
        return m.to(torch.bfloat16)

    m_ref, m_off = build(), build()
    opt_ref = FlatAdamW(m_ref, lr=1e-2, weight_decay=0.01)
    opt_off = FlatAdamW(m_off, lr=1e-2, weight_decay=0.01, offload=True)
    assert all(g.offload for g in opt_off.groups)
    for _ in range(3):
        x = torch.randn(4, 16).to(torch.bfloat16)
        for m, opt in ((m_ref, opt_ref), (m_off, opt_off)):
            opt.zero_grad()
            m(x).float().pow(2).mean().backward()
            opt.step()
    for pr, po in zip(m_ref.parameters(), m_off.parameters()):
        torch.testing.assert_close(pr, po)
    # state dicts round-trip regardless of placement
    sd = opt_off.state_dict()
    opt_ref.load_state_dict(sd)


def test_flat_adamw_offload_fp32_unsharded_falls_back():
    """fp32 world==1 params alias the master: offload must be refused
    silently (the model itself cannot leave the device)."""
    import torch.nn as nn
    from luminaai_amd.training.optimizer import FlatAdamW
    m = nn.Linear(8, 8)
    opt = FlatAdamW(m, lr=1e-3, offload=True)
    assert all(not g.offload for g in opt.groups)
    opt.zero_grad()
    m(torch.randn(2, 8)).sum().backward()
    opt.step()  # still steps fine


def test_pad_k_cache_invalidation():
    """_pad_k_cached serves stale data only until invalidate_pad_cache();
    the Trainer calls it every optimizer step (grad_x correctness)."""
    import torch
    from luminaai_amd.ops.interface import (_pad_k_cached,
                                            invalidate_pad_cache)
    invalidate_pad_cache()
    w = torch.randn(2, 4, 6)
    p1 = _pad_k_cached(w, 2)
    assert p1.shape == (2, 4, 8)
    assert torch.equal(p1[..., :6], w) and p1[..., 6:].abs().sum() == 0
    assert _pad_k_cached(w, 2) is p1          # cache hit
    with torch.no_grad():
        w.mul_(2.0)                           # in-place update...
    invalidate_pad_cache()                    # ...trainer invalidates
    p2 = _pad_k_cached(w, 2)
    assert torch.equal(p2[..., :6], w)
    # id-recycling safety: a NEW tensor at a recycled id must miss
    del w
    w2 = torch.randn(2, 4, 6)
    p3 = _pad_k_cached(w2, 2)
    assert torch.equal(p3[..., :6], w2)
