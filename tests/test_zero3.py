"""ZeRO-3 engine tests: single-process equivalence with the ZeRO-0 path
(fp32, CPU) and multi-process (gloo x2) sync + DDP equivalence."""

import os
import socket

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2


def _cfg(zero_stage, rank=0, ckpt=False):
    from luminaai_amd.config import Config
    return Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                  num_kv_heads=2, seq_length=32, intermediate_size=128,
                  micro_batch_size=2, gradient_accumulation_steps=1,
                  num_workers=0, use_moe=False, use_mod=False,
                  zero_stage=zero_stage, precision="fp32",
                  gradient_checkpointing=ckpt,
                  experiment_name=f"z3_test_{zero_stage}_r{rank}",
                  eval_every_n_batches=0, save_every_n_batches=0)


def _train(cfg, steps=3, seed=500):
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.training import Trainer
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    torch.manual_seed(seed)
    for _ in range(steps):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        t.engine.set_sync(True)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    return t


def _full_weights(t):
    """Materialise and snapshot all params."""
    if t.engine.stage >= 3:
        with t.engine.gathered_weights():
            return {k: v.detach().clone()
                    for k, v in t.model.state_dict().items()}
    return {k: v.detach().clone() for k, v in t.model.state_dict().items()}


def test_zero3_matches_zero0_single_proc():
    t0 = _train(_cfg(0))
    t3 = _train(_cfg(3))
    w0 = _full_weights(t0)
    w3 = _full_weights(t3)
    for k in w0:
        torch.testing.assert_close(w3[k], w0[k], rtol=1e-5, atol=1e-6,
                                   msg=f"mismatch in {k}")
    assert t3.optimizer.last_grad_norm() == pytest.approx(
        t0.optimizer.last_grad_norm(), rel=1e-5)


def test_zero3_with_activation_checkpointing():
    t_plain = _train(_cfg(3, ckpt=False))
    t_ckpt = _train(_cfg(3, ckpt=True))
    wp = _full_weights(t_plain)
    wc = _full_weights(t_ckpt)
    for k in wp:
        torch.testing.assert_close(wc[k], wp[k], rtol=1e-5, atol=1e-6,
                                   msg=f"mismatch in {k}")


def test_zero3_offloaded_matches_resident():
    """cpu_offload_optimizer=True under ZeRO-3: host-resident master/m/v
    and the two-phase CPU step must train identically to the resident
    path (same RNG, same data)."""
    cfg = _cfg(3)
    cfg.cpu_offload_optimizer = True
    t_off = _train(cfg)
    assert all(seg.offload for u in t_off.engine.units for seg in u.segments)
    t_res = _train(_cfg(3))
    wo = _full_weights(t_off)
    wr = _full_weights(t_res)
    for k in wr:
        torch.testing.assert_close(wo[k], wr[k], rtol=1e-5, atol=1e-6,
                                   msg=f"mismatch in {k}")
    assert t_off.optimizer.last_grad_norm() == pytest.approx(
        t_res.optimizer.last_grad_norm(), rel=1e-5)


def test_zero3_grad_accumulation():
    cfg = _cfg(3)
    cfg.gradient_accumulation_steps = 2
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.training import Trainer
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    torch.manual_seed(42)
    for _ in range(2):  # two micro-batches
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    t.optimizer_step()
    assert t.global_step == 1
    assert t.optimizer.last_grad_norm() > 0


def test_zero3_checkpoint_roundtrip(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    t = _train(_cfg(3), steps=2)
    path = t.save_checkpoint()
    assert path and os.path.exists(path)
    w_before = _full_weights(t)
    # train further, then roll back
    torch.manual_seed(99)
    ids = torch.randint(1, 512, (2, 33))
    t.engine.set_sync(True)
    t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    t.optimizer_step()
    t.load_checkpoint(path)
    w_after = _full_weights(t)
    for k in w_before:
        torch.testing.assert_close(w_after[k], w_before[k],
                                   msg=f"mismatch in {k}")


# ---------------------------------------------------------------- multi-proc
def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _run(rank, world, port, fn_name, q):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
    })
    dist.init_process_group("gloo", init_method="env://", rank=rank,
                            world_size=world)
    try:
        import test_zero3
        result = getattr(test_zero3, fn_name)(rank, world)
        q.put((rank, "ok", result))
    except Exception:  # noqa: BLE001
        import traceback
        q.put((rank, "err", traceback.format_exc()))
    finally:
        dist.destroy_process_group()


def _spawn(fn_name):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_run, args=(r, WORLD, port, fn_name, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(WORLD):
        rank, status, payload = q.get()
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=60)
    return results


def zero3_worker(rank, world):
    t = _train(_cfg(3, rank=rank), seed=500 + rank)
    w = _full_weights(t)
    key = "embed_tokens.weight"
    return {"checksum": float(w[key].sum()), "norm": float(w[key].norm()),
            "grad_norm": t.optimizer.last_grad_norm()}


def ddp_ref_worker(rank, world):
    t = _train(_cfg(0, rank=rank), seed=500 + rank)
    w = _full_weights(t)
    key = "embed_tokens.weight"
    return {"checksum": float(w[key].sum()), "norm": float(w[key].norm()),
            "grad_norm": t.optimizer.last_grad_norm()}


def test_zero3_ranks_in_sync_and_match_ddp():
    z3 = _spawn("zero3_worker")
    assert z3[0]["checksum"] == pytest.approx(z3[1]["checksum"], abs=1e-4)
    ddp = _spawn("ddp_ref_worker")
    assert z3[0]["checksum"] == pytest.approx(ddp[0]["checksum"], rel=1e-4)
    assert z3[0]["grad_norm"] == pytest.approx(ddp[0]["grad_norm"], rel=1e-4)


def zero3_save_worker(rank, world):
    """Multi-rank ZeRO-3 save must not deadlock (gather is collective) and
    must write rank-0 main file + rank-1 optimizer shard."""
    from luminaai_amd.training import CheckpointManager
    tmp = os.environ["Z3_SAVE_TMP"]
    t = _train(_cfg(3, rank=rank), steps=1, seed=500 + rank)
    t.checkpoints = CheckpointManager(os.path.join(tmp, "z3_ckpts"))
    path = t.save_checkpoint()
    dist.barrier()
    files = sorted(os.listdir(os.path.join(tmp, "z3_ckpts")))
    return {"path": path, "files": files}


def test_zero3_multirank_save(tmp_path):
    os.environ["Z3_SAVE_TMP"] = str(tmp_path)
    res = _spawn("zero3_save_worker")
    assert res[0]["path"].endswith(".pt")
    assert res[1]["path"] == ""          # rank1 writes only its shard
    names = res[0]["files"]
    assert any(n.startswith("checkpoint_step_") for n in names), names
    assert any(n.startswith("optim_shard_") and "rank1" in n
               for n in names), names


def test_zero3_hybrid_moe_mod_matches_zero0():
    """ZeRO-3 over the hybrid MoE+MoD model (the ~70B preset's structure)
    must train identically to ZeRO-0."""
    def cfg(stage):
        from luminaai_amd.config import Config
        return Config(vocab_size=512, hidden_size=64, num_layers=4,
                      num_heads=4, num_kv_heads=2, seq_length=32,
                      intermediate_size=128, micro_batch_size=2,
                      gradient_accumulation_steps=1, num_workers=0,
                      use_moe=True, num_experts=4, moe_top_k=2,
                      routing_noise_std=0.0, moe_pattern="every_2nd",
                      use_mod=True, mod_capacity_factor=0.5,
                      zero_stage=stage, precision="fp32",
                      experiment_name=f"z3h_{stage}",
                      eval_every_n_batches=0, save_every_n_batches=0)

    t0 = _train(cfg(0), steps=2)
    t3 = _train(cfg(3), steps=2)
    w0 = _full_weights(t0)
    w3 = _full_weights(t3)
    for k in w0:
        torch.testing.assert_close(w3[k], w0[k], rtol=1e-5, atol=1e-6,
                                   msg=f"mismatch in {k}")


# ---- ZeRO-3 + expert parallelism -------------------------------------------
def z3_ep_worker(rank, world):
    """ZeRO-3 composed with EP must train identically to ZeRO-0 + EP:
    dense params shard over the world, expert params stay local to their
    EP rank (replica group of one under full EP)."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.training import Trainer

    def run(stage):
        reset_mesh()
        cfg = Config(vocab_size=512, hidden_size=64, num_layers=2,
                     num_heads=4, num_kv_heads=2, seq_length=32,
                     intermediate_size=128, micro_batch_size=2,
                     gradient_accumulation_steps=1, num_workers=0,
                     use_moe=True, num_experts=4, moe_top_k=2,
                     routing_noise_std=0.0, use_mod=False,
                     zero_stage=stage, precision="fp32",
                     experiment_name=f"z3ep_s{stage}_r{rank}",
                     eval_every_n_batches=0, save_every_n_batches=0,
                     gradient_checkpointing=False)
        init_mesh(world)
        torch.manual_seed(1234)
        model = DeepSeekTransformer(config_to_deepseek_config(cfg))
        t = Trainer(model, ConversationTokenizer(), cfg)
        t._setup_scheduler(10)
        torch.manual_seed(700 + rank)
        for _ in range(2):
            ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
            t.engine.set_sync(True)
            t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
            t.optimizer_step()
        if stage >= 3:
            with t.engine.gathered_weights():
                dense = t.model.embed_tokens.weight.detach().clone()
                exp = t.model.layers[0].ffn.w_gate_up.detach().clone()
        else:
            dense = t.model.embed_tokens.weight.detach().clone()
            exp = t.model.layers[0].ffn.w_gate_up.detach().clone()
        gn = t.optimizer.last_grad_norm()
        if hasattr(t.engine, "remove_hooks"):
            t.engine.remove_hooks()
        return dense, exp, gn

    d0, e0, gn0 = run(0)
    d3, e3, gn3 = run(3)
    return {
        "dense_err": float((d0 - d3).abs().max()),
        "exp_err": float((e0 - e3).abs().max()),
        "gn0": gn0, "gn3": gn3,
        "exp_sum": float(e3.sum()),
    }


def test_zero3_with_expert_parallelism():
    res = _spawn("z3_ep_worker")
    for r in range(WORLD):
        assert res[r]["dense_err"] < 1e-4, res
        assert res[r]["exp_err"] < 1e-4, res
        assert res[r]["gn3"] == pytest.approx(res[r]["gn0"], rel=1e-4), res
    # different EP ranks hold different experts
    assert res[0]["exp_sum"] != res[1]["exp_sum"]


def test_zero3_expert_add_prune():
    """Expert add/prune under ZeRO-3 (round-2 roadmap item 9): the engine
    is rebuilt over the mutated shapes and training continues."""
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.training import Trainer
    cfg = _cfg(3)
    cfg.use_moe = True
    cfg.num_experts = 4
    cfg.moe_top_k = 2
    cfg.routing_noise_std = 0.0
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(20)

    def _steps(n, seed):
        torch.manual_seed(seed)
        losses = []
        for _ in range(n):
            ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
            out = t.train_step({"input_ids": ids[:, :-1],
                                "labels": ids[:, 1:]})
            t.optimizer_step()
            losses.append(out["loss"])
        return losses

    _steps(2, 900)
    sc_before = t.optimizer.step_count
    assert t.add_expert()
    assert t.model.layers[0].ffn.num_experts == 5
    assert t.optimizer.step_count == sc_before     # bias correction carries
    ls = _steps(2, 901)
    assert all(torch.isfinite(torch.tensor(ls))), ls
    # grown gate rows are optimizable (w_gate_up present in a segment)
    segs = [seg for u in t.engine.units for seg in u.segments]
    n_flat = sum(seg.numel for seg in segs)
    n_model = sum(p.numel() for p in t.model.parameters())
    assert n_flat == n_model

    assert t.prune_expert()
    assert t.model.layers[0].ffn.num_experts == 4
    ls = _steps(2, 902)
    assert all(torch.isfinite(torch.tensor(ls))), ls
    n_flat = sum(seg.numel
                 for u in t.engine.units for seg in u.segments)
    assert n_flat == sum(p.numel() for p in t.model.parameters())
    t.engine.remove_hooks()


def test_zero3_rejects_tp_mesh():
    """ZeRO-3 + TP is not a supported composition (the dense-segment
    collectives assume the world group is pure DP); it must raise a
    clear error instead of deadlocking mid-training."""
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.parallel.zero3 import Zero3Engine
    from luminaai_amd.parallel import mesh as mesh_mod
    cfg = _cfg(3)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))

    class _FakeMesh:
        tp_size = 2
        ep_size = 1

    with pytest.raises(ValueError, match="ZeRO-3.*TP"):
        Zero3Engine(model, cfg, mesh=_FakeMesh())


def test_zero3_expert_load_balance():
    """apply_expert_load_balance under ZeRO-3 routes through the gather->
    re-place->rebuild path (single-proc, ep==1 means nothing to balance;
    force a placement by faking ep metadata on the layer is out of scope
    -- this validates the path does not corrupt training when no rebuild
    is needed and the Z3 gate no longer short-circuits)."""
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.training import Trainer
    cfg = _cfg(3)
    cfg.use_moe = True
    cfg.num_experts = 4
    cfg.moe_top_k = 2
    cfg.routing_noise_std = 0.0
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(20)
    torch.manual_seed(905)
    for _ in range(2):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    assert t.apply_expert_load_balance() is False   # ep==1: no-op
    out = t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    assert torch.isfinite(torch.tensor(out["loss"]))
    t.engine.remove_hooks()


def z3_elastic_save_worker(rank, world):
    """Train ZeRO-3 at world=2 with SAME data on both ranks and save."""
    from luminaai_amd.training import CheckpointManager
    tmp = os.environ["Z3_ELASTIC_TMP"]
    os.chdir(tmp)
    cfg = _cfg(3, rank=rank)
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.training import Trainer
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    t.checkpoints = CheckpointManager(os.path.join(tmp, "shared_ckpts"))
    torch.manual_seed(750)
    for _ in range(2):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        t.engine.set_sync(True)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    t.save_checkpoint()
    dist.barrier()
    m_sum = sum(float(g.m.sum()) for g in t.optimizer.groups)
    t.engine.remove_hooks()
    return {"m_sum": m_sum, "steps": t.optimizer.step_count}


def test_zero3_elastic_resume_world1(tmp_path):
    os.environ["Z3_ELASTIC_TMP"] = str(tmp_path)
    res = _spawn("z3_elastic_save_worker")
    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        from luminaai_amd.training import CheckpointManager
        # reference arm: identical run entirely in this process at world=1
        t_ref = _train(_cfg(3, rank=9), steps=2, seed=750)
        # elastic arm: fresh world-1 trainer resuming the world-2 save
        from luminaai_amd.data.tokenizer import ConversationTokenizer
        from luminaai_amd.models import (DeepSeekTransformer,
                                         config_to_deepseek_config)
        from luminaai_amd.training import Trainer
        cfg = _cfg(3, rank=8)
        torch.manual_seed(1234)
        model = DeepSeekTransformer(config_to_deepseek_config(cfg))
        t = Trainer(model, ConversationTokenizer(), cfg)
        t._setup_scheduler(10)
        t.checkpoints = CheckpointManager(str(tmp_path / "shared_ckpts"))
        t.load_checkpoint("latest")
        assert t.optimizer.step_count == res[0]["steps"]
        m_here = sum(float(g.m.sum()) for g in t.optimizer.groups)
        assert m_here == pytest.approx(res[0]["m_sum"] + res[1]["m_sum"],
                                       rel=1e-5)
        # resumed state matches the uninterrupted world-1 reference
        w_ref = _full_weights(t_ref)
        w_new = _full_weights(t)
        for k in w_ref:
            torch.testing.assert_close(w_new[k], w_ref[k], rtol=1e-4,
                                       atol=1e-5, msg=k)
    finally:
        os.chdir(cwd)


def z3_ep_subworld_worker(rank, world):
    """world=4, ep=2: expert replica groups have TWO members, so expert
    segments really shard (the ep==world case degenerates to local
    state). ZeRO-3 must still match ZeRO-0 on the same mesh."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.training import Trainer

    def run(stage):
        reset_mesh()
        cfg = Config(vocab_size=512, hidden_size=64, num_layers=2,
                     num_heads=4, num_kv_heads=2, seq_length=32,
                     intermediate_size=128, micro_batch_size=2,
                     gradient_accumulation_steps=1, num_workers=0,
                     use_moe=True, num_experts=4, moe_top_k=2,
                     routing_noise_std=0.0, use_mod=False,
                     zero_stage=stage, precision="fp32",
                     experiment_name=f"z3ep2_s{stage}_r{rank}",
                     eval_every_n_batches=0, save_every_n_batches=0,
                     gradient_checkpointing=False)
        init_mesh(ep_size=2)
        torch.manual_seed(1234)
        model = DeepSeekTransformer(config_to_deepseek_config(cfg))
        t = Trainer(model, ConversationTokenizer(), cfg)
        t._setup_scheduler(10)
        torch.manual_seed(800 + rank)
        for _ in range(2):
            ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
            t.engine.set_sync(True)
            t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
            t.optimizer_step()
        if stage >= 3:
            with t.engine.gathered_weights():
                dense = t.model.embed_tokens.weight.detach().clone()
                exp = t.model.layers[1].ffn.w_gate_up.detach().clone()
        else:
            dense = t.model.embed_tokens.weight.detach().clone()
            exp = t.model.layers[1].ffn.w_gate_up.detach().clone()
        if hasattr(t.engine, "remove_hooks"):
            t.engine.remove_hooks()
        return dense, exp

    d0, e0 = run(0)
    d3, e3 = run(3)
    return {"dense_err": float((d0 - d3).abs().max()),
            "exp_err": float((e0 - e3).abs().max())}


def test_zero3_ep_subworld():
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_run, args=(r, 4, port,
                                            "z3_ep_subworld_worker", q))
             for r in range(4)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(4):
        rank, status, payload = q.get()
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        results[rank] = payload
    for p in procs:
        p.join(timeout=180)
    for r in range(4):
        assert results[r]["dense_err"] < 1e-4, results
        assert results[r]["exp_err"] < 1e-4, results
