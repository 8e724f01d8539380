"""Ring attention (context parallelism): forward AND gradients must equal
full attention over the global sequence (gloo x2 and x4)."""

import os
import socket

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _ref(q, k, v, causal, scale):
    scores = torch.matmul(q, k.transpose(-1, -2)) * scale
    if causal:
        S = scores.shape[-1]
        cm = torch.ones(S, S, dtype=torch.bool).tril()
        scores = scores.masked_fill(~cm, float("-inf"))
    return torch.softmax(scores, dim=-1) @ v


def ring_worker(rank, world, causal):
    from luminaai_amd.parallel.context_parallel import ring_attention
    B, H, S, D = 2, 3, 8 * world, 16
    Sl = S // world
    torch.manual_seed(99)
    q = torch.randn(B, H, S, D, requires_grad=True)
    k = torch.randn(B, H, S, D, requires_grad=True)
    v = torch.randn(B, H, S, D, requires_grad=True)
    torch.manual_seed(100)
    gout = torch.randn(B, H, S, D)
    scale = D ** -0.5

    ref = _ref(q, k, v, causal, scale)
    ref.backward(gout)

    lo, hi = rank * Sl, (rank + 1) * Sl
    ql = q.detach()[:, :, lo:hi].clone().requires_grad_(True)
    kl = k.detach()[:, :, lo:hi].clone().requires_grad_(True)
    vl = v.detach()[:, :, lo:hi].clone().requires_grad_(True)
    out = ring_attention(ql, kl, vl, causal=causal)
    out.backward(gout[:, :, lo:hi])

    def err(a, b):
        return float((a - b).abs().max())

    return {
        "out": err(out, ref[:, :, lo:hi]),
        "dq": err(ql.grad, q.grad[:, :, lo:hi]),
        "dk": err(kl.grad, k.grad[:, :, lo:hi]),
        "dv": err(vl.grad, v.grad[:, :, lo:hi]),
    }


def _run(rank, world, port, causal, q):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
    })
    dist.init_process_group("gloo", init_method="env://", rank=rank,
                            world_size=world)
    try:
        q.put((rank, "ok", ring_worker(rank, world, causal)))
    except Exception:  # noqa: BLE001
        import traceback
        q.put((rank, "err", traceback.format_exc()))
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world,causal", [(2, True), (2, False), (4, True)])
def test_ring_attention_matches_full(world, causal):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_run, args=(r, world, port, causal, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, status, payload = q.get()
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=120)
    for r in range(world):
        for key, e in results[r].items():
            assert e < 1e-4, (r, key, results)


def test_ring_attention_single_process():
    """world=1 degenerates to plain (causal) attention."""
    from luminaai_amd.parallel.context_parallel import ring_attention
    torch.manual_seed(0)
    B, H, S, D = 2, 2, 16, 8
    q = torch.randn(B, H, S, D, requires_grad=True)
    k = torch.randn(B, H, S, D, requires_grad=True)
    v = torch.randn(B, H, S, D, requires_grad=True)
    out = ring_attention(q, k, v, causal=True)
    ref = _ref(q.detach(), k.detach(), v.detach(), True, D ** -0.5)
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
    out.sum().backward()
    assert q.grad is not None and torch.isfinite(q.grad).all()
