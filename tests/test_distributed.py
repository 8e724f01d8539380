"""Multi-process (gloo, world=2) tests of the native DP/ZeRO engine.

Pattern: spawn N local processes with a free-port rendezvous (the approach the
vendored ColossalAI testing/utils.py:212-260 uses — re-implemented here)."""

import os
import socket

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2


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
        result = globals()[fn_name](rank, world)
        q.put((rank, "ok", result))
    except Exception as e:  # noqa: BLE001
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


# ---- worker bodies --------------------------------------------------------
def _make_trainer(rank, world, zero_stage):
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.training import Trainer
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=False, use_mod=False,
                 zero_stage=zero_stage, precision="fp32",
                 experiment_name=f"dist_test_r{rank}",
                 eval_every_n_batches=0, save_every_n_batches=0)
    torch.manual_seed(1234)  # same init on all ranks
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    return t, cfg


def _train_worker(rank, world, zero_stage):
    t, cfg = _make_trainer(rank, world, zero_stage)
    torch.manual_seed(500 + rank)  # different data per rank
    for _ in range(3):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        batch = {"input_ids": ids[:, :-1], "labels": ids[:, 1:]}
        t.engine.set_sync(True)
        t.train_step(batch)
        t.optimizer_step()
    w = t.model.embed_tokens.weight.detach()
    return {"checksum": float(w.sum()), "norm": float(w.norm())}


def ddp_worker(rank, world):
    return _train_worker(rank, world, zero_stage=0)


def zero1_worker(rank, world):
    return _train_worker(rank, world, zero_stage=1)


def zero2_worker(rank, world):
    return _train_worker(rank, world, zero_stage=2)


def singleproc_reference(zero_stage=0):
    """Same total batch on ONE process: DP result must match."""
    os.environ.pop("WORLD_SIZE", None)
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.training import Trainer
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=4, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=False, use_mod=False, zero_stage=0,
                 precision="fp32", experiment_name="dist_ref",
                 eval_every_n_batches=0, save_every_n_batches=0)
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    gens = [torch.Generator().manual_seed(500 + r) for r in range(WORLD)]
    for _ in range(3):
        rows = []
        for g in gens:
            ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1),
                                generator=g)
            rows.append(ids)
        ids = torch.cat(rows)
        batch = {"input_ids": ids[:, :-1], "labels": ids[:, 1:]}
        t.train_step(batch)
        t.optimizer_step()
    w = t.model.embed_tokens.weight.detach()
    return {"checksum": float(w.sum()), "norm": float(w.norm())}


# ---- expert parallelism ---------------------------------------------------
def _moe_model_cfg():
    from luminaai_amd.models.transformer import DeepSeekConfig
    return DeepSeekConfig(
        vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
        num_kv_heads=2, intermediate_size=128, seq_length=32,
        use_moe=True, num_experts=4, moe_top_k=2, routing_noise_std=0.0,
        moe_pattern="all", use_mod=False)


def ep_forward_backward_worker(rank, world):
    """EP(2) MoE forward must equal the single-rank full-expert model on the
    same weights; expert grads under EP must equal the SUM of both ranks'
    full-model grads (each expert's grad gathers every rank's tokens)."""
    import torch.distributed as dist
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh

    mcfg = _moe_model_cfg()
    reset_mesh()
    torch.manual_seed(1234)
    full = DeepSeekTransformer(mcfg)            # all 4 experts local

    init_mesh(world)                            # ep == world == 2
    torch.manual_seed(1234)
    ep = DeepSeekTransformer(mcfg)              # 2 local experts
    EL = mcfg.num_experts // world
    with torch.no_grad():
        fp = dict(full.named_parameters())
        for name, p in ep.named_parameters():
            src = fp[name]
            if p.shape != src.shape:            # expert shard
                p.copy_(src[rank * EL:(rank + 1) * EL])
            else:
                p.copy_(src)

    torch.manual_seed(600 + rank)               # different tokens per rank
    ids = torch.randint(1, mcfg.vocab_size, (2, mcfg.seq_length))
    full.train(); ep.train()

    logits_f, aux_f, _ = full(ids)
    logits_e, aux_e, _ = ep(ids)
    torch.testing.assert_close(logits_e, logits_f, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(aux_e, aux_f, rtol=1e-4, atol=1e-5)

    (logits_f.float().pow(2).mean() + aux_f).backward()
    (logits_e.float().pow(2).mean() + aux_e).backward()

    # dense param grads (local, unreduced) must match the full model's
    ge = dict(ep.named_parameters())["layers.0.attention.qkv_proj.weight"].grad
    gf = dict(full.named_parameters())["layers.0.attention.qkv_proj.weight"].grad
    torch.testing.assert_close(ge, gf, rtol=1e-3, atol=1e-5)

    # expert grads: EP grad == sum over ranks of full-model grads (shard slice)
    gf_exp = dict(full.named_parameters())["layers.0.ffn.w_gate_up"].grad.clone()
    dist.all_reduce(gf_exp)
    ge_exp = dict(ep.named_parameters())["layers.0.ffn.w_gate_up"].grad
    torch.testing.assert_close(ge_exp, gf_exp[rank * EL:(rank + 1) * EL],
                               rtol=1e-3, atol=1e-5)
    reset_mesh()
    return {"ok": True}


def ep_train_worker(rank, world):
    """A full EP training step through Trainer + ZeroEngine: dense params
    stay identical across ranks; expert shards evolve independently."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.training import Trainer
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=True, num_experts=4, moe_top_k=2,
                 routing_noise_std=0.0, use_mod=False,
                 zero_stage=0, precision="fp32",
                 experiment_name=f"ep_test_r{rank}",
                 eval_every_n_batches=0, save_every_n_batches=0)
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
    dense = t.model.embed_tokens.weight.detach()
    exp = t.model.layers[0].ffn.w_gate_up.detach()
    reset_mesh()
    return {"dense_sum": float(dense.sum()), "exp_sum": float(exp.sum()),
            "grad_norm": t.optimizer.last_grad_norm()}


def test_ep_forward_backward_equivalence():
    res = _spawn("ep_forward_backward_worker")
    assert res[0]["ok"] and res[1]["ok"]


def test_ep_training_step():
    res = _spawn("ep_train_worker")
    # dense params DP-synced; expert shards differ (different experts)
    assert res[0]["dense_sum"] == pytest.approx(res[1]["dense_sum"], abs=1e-4)
    assert res[0]["exp_sum"] != res[1]["exp_sum"]
    # clip norm agreed across ranks (it is a collective)
    assert res[0]["grad_norm"] == pytest.approx(res[1]["grad_norm"], rel=1e-5)


# ---- Ulysses sequence parallelism -----------------------------------------
def sp_forward_backward_worker(rank, world):
    """SP(2) forward must equal the full-sequence single-rank model on this
    rank's sequence slice; grads of replicated params must sum to the
    full-model grads."""
    import torch.distributed as dist
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.models.transformer import DeepSeekConfig
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh

    mcfg = DeepSeekConfig(vocab_size=512, hidden_size=64, num_layers=2,
                          num_heads=4, num_kv_heads=2, intermediate_size=128,
                          seq_length=64, use_moe=False, use_mod=False)
    reset_mesh()
    torch.manual_seed(1234)
    full = DeepSeekTransformer(mcfg)

    init_mesh(sp_size=world)
    torch.manual_seed(1234)
    sp_model = DeepSeekTransformer(mcfg)
    with torch.no_grad():
        for p_sp, p_f in zip(sp_model.parameters(), full.parameters()):
            p_sp.copy_(p_f)

    torch.manual_seed(777)  # SAME full batch on every rank
    ids = torch.randint(1, mcfg.vocab_size, (2, 64))
    S_loc = 64 // world
    ids_loc = ids[:, rank * S_loc:(rank + 1) * S_loc]

    logits_f, _, _ = full(ids)
    logits_sp, _, _ = sp_model(ids_loc)
    torch.testing.assert_close(
        logits_sp, logits_f[:, rank * S_loc:(rank + 1) * S_loc],
        rtol=2e-4, atol=2e-4)

    # backward: mean over LOCAL tokens; sum of rank grads / world must equal
    # the full-model grad of the global-mean loss
    logits_f.float().pow(2).mean().backward()
    logits_sp.float().pow(2).mean().backward()
    g_sp = dict(sp_model.named_parameters())[
        "layers.0.attention.qkv_proj.weight"].grad.clone()
    dist.all_reduce(g_sp)
    g_sp /= world
    g_f = dict(full.named_parameters())[
        "layers.0.attention.qkv_proj.weight"].grad
    # different GEMM decompositions reorder the fp32 reductions: compare
    # with an absolute tolerance scaled to the grad magnitude
    torch.testing.assert_close(g_sp, g_f, rtol=0.05,
                               atol=1e-4 * g_f.abs().max().item())
    reset_mesh()
    return {"ok": True}


def test_ulysses_sp_equivalence():
    res = _spawn("sp_forward_backward_worker")
    assert res[0]["ok"] and res[1]["ok"]


# ---- tests ---------------------------------------------------------------
@pytest.mark.parametrize("worker", ["ddp_worker", "zero1_worker", "zero2_worker"])
def test_ranks_stay_in_sync(worker):
    res = _spawn(worker)
    assert res[0]["checksum"] == pytest.approx(res[1]["checksum"], abs=1e-4)
    assert res[0]["norm"] == pytest.approx(res[1]["norm"], abs=1e-4)


def test_ddp_matches_single_process():
    """2-rank DP with per-rank micro-batch 2 == 1-process batch 4.
    (loss is per-token mean within micro-batch; DP averages rank means, the
    single-process run averages over the combined batch — identical here
    because every row has the same token count)."""
    dist_res = _spawn("ddp_worker")
    ref = singleproc_reference()
    assert dist_res[0]["checksum"] == pytest.approx(ref["checksum"], rel=1e-4)


def test_zero2_matches_ddp():
    a = _spawn("ddp_worker")
    b = _spawn("zero2_worker")
    assert a[0]["checksum"] == pytest.approx(b[0]["checksum"], rel=1e-4)


# ---- expert load balancing -------------------------------------------------
def ep_load_balance_worker(rank, world):
    """Applying a non-trivial placement must not change model outputs."""
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.parallel.load_balance import apply_placement
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh

    mcfg = _moe_model_cfg()
    init_mesh(world)
    torch.manual_seed(1234)
    ep = DeepSeekTransformer(mcfg).eval()
    torch.manual_seed(42 + rank)
    ids = torch.randint(1, mcfg.vocab_size, (2, mcfg.seq_length))
    with torch.no_grad():
        before, _, _ = ep(ids)
        order = [1, 3, 0, 2]        # shuffle experts across both ranks
        for l in ep.get_moe_layers():
            apply_placement(l, order)
        after, _, _ = ep(ids)
        # and back to identity
        for l in ep.get_moe_layers():
            apply_placement(l, list(range(mcfg.num_experts)))
        back, _, _ = ep(ids)
    reset_mesh()
    d1 = float((after - before).abs().max())
    d2 = float((back - before).abs().max())
    return {"d_placed": d1, "d_back": d2}


def test_ep_load_balance_placement_transparent():
    res = _spawn("ep_load_balance_worker")
    for r in range(2):
        assert res[r]["d_placed"] < 1e-4, res
        assert res[r]["d_back"] < 1e-4, res


def test_plan_placement_lpt():
    from luminaai_amd.parallel.load_balance import imbalance, plan_placement
    loads = [0.4, 0.3, 0.1, 0.05, 0.05, 0.04, 0.03, 0.03]
    order = plan_placement(loads, 4)
    assert sorted(order) == list(range(8))
    assert imbalance(loads, order, 4) < imbalance(loads, list(range(8)), 4)
    # the two heaviest experts must land on different ranks
    pos = {e: i // 2 for i, e in enumerate(order)}
    assert pos[0] != pos[1]


# ---- ZeRO shard checkpoint resume ------------------------------------------
def zero2_ckpt_resume_worker(rank, world):
    """Save under ZeRO-2, train on, reload: every rank's optimizer shard
    must restore and training must continue identically to an uninterrupted
    run."""
    from luminaai_amd.training import CheckpointManager
    tmp = os.environ["Z2_CKPT_TMP"]
    os.chdir(tmp)
    t, cfg = _make_trainer(rank, world, zero_stage=2)
    t.checkpoints = CheckpointManager(os.path.join(tmp, "shared_ckpts"))
    torch.manual_seed(500 + rank)
    batches = []
    for _ in range(4):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        batches.append({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    for b in batches[:2]:
        t.engine.set_sync(True)
        t.train_step(b)
        t.optimizer_step()
    path = t.save_checkpoint()
    dist.barrier()
    # continue training (the "uninterrupted" arm)
    for b in batches[2:]:
        t.engine.set_sync(True)
        t.train_step(b)
        t.optimizer_step()
    w_cont = float(t.model.embed_tokens.weight.detach().sum())
    m_cont = float(t.optimizer.groups[0].m.sum())

    # fresh trainer, resume from the checkpoint, replay the same batches
    t2, _ = _make_trainer(rank, world, zero_stage=2)
    t2.checkpoints = t.checkpoints
    t2.load_checkpoint("latest" if rank != 0 else (path or "latest"))
    assert t2.global_step == 2
    for b in batches[2:]:
        t2.engine.set_sync(True)
        t2.train_step(b)
        t2.optimizer_step()
    w_res = float(t2.model.embed_tokens.weight.detach().sum())
    m_res = float(t2.optimizer.groups[0].m.sum())
    return {"w_cont": w_cont, "w_res": w_res, "m_cont": m_cont, "m_res": m_res}


def test_zero2_checkpoint_resume_equivalence(tmp_path):
    os.environ["Z2_CKPT_TMP"] = str(tmp_path)
    res = _spawn("zero2_ckpt_resume_worker")
    for r in range(2):
        assert res[r]["w_res"] == pytest.approx(res[r]["w_cont"], rel=1e-5), res
        assert res[r]["m_res"] == pytest.approx(res[r]["m_cont"], rel=1e-4), res


# ---- data sharding ---------------------------------------------------------
def dp_sampler_worker(rank, world):
    """DP ranks must receive disjoint data slices."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.dataset import SyntheticDataset, create_dataloader
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=16, micro_batch_size=2,
                 num_workers=0, use_moe=False, use_mod=False,
                 experiment_name=f"smp_{rank}", seed=7)
    ds = SyntheticDataset(cfg.vocab_size, cfg.seq_length, 16, seed=7)
    dl = create_dataloader(ds, cfg, shuffle=True)
    first = next(iter(dl))
    assert len(dl) == 16 // 2 // world     # per-rank slice
    return {"sum": float(first["input_ids"].sum())}


def test_dp_data_sharding():
    res = _spawn("dp_sampler_worker")
    assert res[0]["sum"] != res[1]["sum"], "ranks received identical data"


# ---- gradless-param bucket flush -------------------------------------------
def gradless_param_worker(rank, world):
    """A parameter that receives no grad (MoD router at capacity 1.0 never
    runs) must not leave its bucket unreduced under overlapped ZeRO-1."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.training import Trainer
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=False, use_mod=True,
                 mod_capacity_factor=1.0,     # run_mod False -> router unused
                 zero_stage=1, precision="fp32",
                 experiment_name=f"gl_{rank}",
                 eval_every_n_batches=0, save_every_n_batches=0)
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    torch.manual_seed(800 + rank)            # different data per rank
    for _ in range(3):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        t.engine.set_sync(True)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    w = torch.cat([p.detach().reshape(-1) for p in t.model.parameters()])
    return {"checksum": float(w.sum())}


def test_gradless_param_bucket_flushed():
    res = _spawn("gradless_param_worker")
    assert res[0]["checksum"] == pytest.approx(res[1]["checksum"], abs=1e-4)


# ---- tensor parallelism ----------------------------------------------------
def tp_forward_backward_worker(rank, world):
    """TP(2) forward must equal the unsharded model; grads of replicated
    params must match after the TP duplicates average out; sharded-weight
    grads must equal the full-model grad slices."""
    import torch.distributed as dist
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.models.transformer import DeepSeekConfig
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.parallel.tensor_parallel import convert_to_tensor_parallel

    mcfg = DeepSeekConfig(vocab_size=512, hidden_size=64, num_layers=2,
                          num_heads=4, num_kv_heads=2, intermediate_size=128,
                          seq_length=32, use_moe=False, use_mod=False)
    reset_mesh()
    torch.manual_seed(1234)
    full = DeepSeekTransformer(mcfg)

    mesh = init_mesh(tp_size=world)
    torch.manual_seed(1234)
    tpm = DeepSeekTransformer(mcfg)
    with torch.no_grad():
        for a, b in zip(tpm.parameters(), full.parameters()):
            a.copy_(b)
    n = convert_to_tensor_parallel(tpm, mesh)
    assert n == 2

    torch.manual_seed(777)            # SAME batch on both TP ranks
    ids = torch.randint(1, mcfg.vocab_size, (2, 32))
    lf, _, _ = full(ids)
    lt, _, _ = tpm(ids)
    torch.testing.assert_close(lt, lf, rtol=2e-4, atol=2e-4)

    lf.float().pow(2).mean().backward()
    lt.float().pow(2).mean().backward()

    # replicated param (embedding): grads identical to the full model's
    ge = tpm.embed_tokens.weight.grad
    gf = full.embed_tokens.weight.grad
    torch.testing.assert_close(ge, gf, rtol=0.05,
                               atol=1e-4 * gf.abs().max().item())

    # sharded down_proj: TP grad == full grad column slice
    li = mcfg.intermediate_size // world
    gt = tpm.layers[0].ffn.down_proj.weight.grad
    gfd = full.layers[0].ffn.down_proj.weight.grad[:, rank * li:(rank + 1) * li]
    torch.testing.assert_close(gt, gfd, rtol=0.05,
                               atol=1e-4 * gfd.abs().max().item())
    reset_mesh()
    return {"ok": True}


def tp_train_worker(rank, world):
    """Trainer + ZeroEngine under TP: replicated params stay in sync,
    sharded params evolve their own slices, one grad norm across ranks."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.parallel.tensor_parallel import convert_to_tensor_parallel
    from luminaai_amd.training import Trainer
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=False, use_mod=False,
                 zero_stage=0, precision="fp32",
                 experiment_name=f"tp_{rank}",
                 eval_every_n_batches=0, save_every_n_batches=0)
    mesh = init_mesh(tp_size=world)
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    convert_to_tensor_parallel(model, mesh)
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    torch.manual_seed(900)            # same data within the TP group
    for _ in range(2):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        t.engine.set_sync(True)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    dense = float(t.model.embed_tokens.weight.detach().sum())
    shard = float(t.model.layers[0].ffn.down_proj.weight.detach().sum())
    reset_mesh()
    return {"dense": dense, "shard": shard,
            "grad_norm": t.optimizer.last_grad_norm()}


def test_tp_forward_backward_equivalence():
    res = _spawn("tp_forward_backward_worker")
    assert res[0]["ok"] and res[1]["ok"]


def test_tp_training_step():
    res = _spawn("tp_train_worker")
    assert res[0]["dense"] == pytest.approx(res[1]["dense"], abs=1e-4)
    assert res[0]["shard"] != res[1]["shard"]
    assert res[0]["grad_norm"] == pytest.approx(res[1]["grad_norm"], rel=1e-4)


def tp_matches_single_worker(rank, world):
    """TP(2) training on a shared batch must produce EXACTLY the same model
    as single-process training on that batch (grad scaling correctness)."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.parallel.tensor_parallel import convert_to_tensor_parallel
    from luminaai_amd.training import Trainer
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=False, use_mod=False,
                 zero_stage=0, precision="fp32",
                 experiment_name=f"tpm_{rank}",
                 eval_every_n_batches=0, save_every_n_batches=0)
    mesh = init_mesh(tp_size=world)
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    convert_to_tensor_parallel(model, mesh)
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    torch.manual_seed(901)
    for _ in range(2):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        t.engine.set_sync(True)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    li = 128 // world
    shard = t.model.layers[0].ffn.down_proj.weight.detach()
    emb = float(t.model.embed_tokens.weight.detach().sum())
    reset_mesh()
    return {"emb": emb, "shard_sum": float(shard.sum()), "rank": rank}


def test_tp_matches_single_process():
    res = _spawn("tp_matches_single_worker")
    # single-process reference with identical seeds/data
    import torch as th
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.training import Trainer
    os.environ.pop("WORLD_SIZE", None)
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=False, use_mod=False,
                 zero_stage=0, precision="fp32", experiment_name="tpm_ref",
                 eval_every_n_batches=0, save_every_n_batches=0)
    th.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    th.manual_seed(901)
    for _ in range(2):
        ids = th.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    emb_ref = float(t.model.embed_tokens.weight.detach().sum())
    dw = t.model.layers[0].ffn.down_proj.weight.detach()
    li = 128 // WORLD
    for r in range(WORLD):
        assert res[r]["emb"] == pytest.approx(emb_ref, abs=1e-3), \
            (res[r]["emb"], emb_ref)
        ref_shard = float(dw[:, r * li:(r + 1) * li].sum())
        assert res[r]["shard_sum"] == pytest.approx(ref_shard, abs=1e-3), \
            (r, res[r]["shard_sum"], ref_shard)


def sp_trainer_worker(rank, world):
    """SP through the Trainer: replicas share batches, each trains on its
    sequence slice; dense params stay in sync."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.training import Trainer
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=64, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=False, use_mod=False,
                 zero_stage=0, precision="fp32",
                 experiment_name=f"sp_t_{rank}",
                 eval_every_n_batches=0, save_every_n_batches=0)
    init_mesh(sp_size=world)
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    torch.manual_seed(910)             # same batch on every SP rank
    for _ in range(2):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        t.engine.set_sync(True)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    w = float(t.model.embed_tokens.weight.detach().sum())
    reset_mesh()
    return {"w": w}


def test_sp_through_trainer():
    res = _spawn("sp_trainer_worker")
    assert res[0]["w"] == pytest.approx(res[1]["w"], abs=1e-4)


# ---- elastic resharding ----------------------------------------------------
def zero1_elastic_save_worker(rank, world):
    """Train 2 steps under ZeRO-1 at world=2 and save; the test's main
    process then resumes the run single-process (world=1)."""
    from luminaai_amd.training import CheckpointManager
    tmp = os.environ["Z1_ELASTIC_TMP"]
    os.chdir(tmp)
    t, cfg = _make_trainer(rank, world, zero_stage=1)
    t.checkpoints = CheckpointManager(os.path.join(tmp, "shared_ckpts"))
    torch.manual_seed(700)         # SAME data on both ranks
    for _ in range(2):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        t.engine.set_sync(True)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    t.save_checkpoint()
    dist.barrier()
    return {"m_saved": float(t.optimizer.groups[0].m.sum()),
            "w_saved": float(t.model.embed_tokens.weight.detach().sum()),
            "steps": t.optimizer.step_count}


def test_zero1_elastic_resume_world1(tmp_path):
    os.environ["Z1_ELASTIC_TMP"] = str(tmp_path)
    res = _spawn("zero1_elastic_save_worker")
    # resume in THIS process at world=1
    import os as _os
    cwd = _os.getcwd()
    _os.chdir(tmp_path)
    try:
        from luminaai_amd.training import CheckpointManager
        t, cfg = _make_trainer(0, 1, zero_stage=1)
        t.checkpoints = CheckpointManager(str(tmp_path / "shared_ckpts"))
        t.load_checkpoint("latest")
        assert t.global_step == 2
        assert t.optimizer.step_count == res[0]["steps"]
        # merged moment mass equals the sum of the saved shards
        m_sum = sum(float(g.m.sum()) for g in t.optimizer.groups)
        m_saved = None
        # per-group shard sums aren't separable from the worker payload;
        # check group 0 against rank sums (padding contributes zeros)
        g0 = float(t.optimizer.groups[0].m.sum())
        assert g0 == pytest.approx(res[0]["m_saved"] + res[1]["m_saved"],
                                   rel=1e-5)
        assert float(t.model.embed_tokens.weight.detach().sum()) == \
            pytest.approx(res[0]["w_saved"], rel=1e-5)
        # training continues
        torch.manual_seed(701)
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        t.engine.set_sync(True)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
        assert torch.isfinite(t.model.embed_tokens.weight.detach()).all()
    finally:
        _os.chdir(cwd)


def zero1_elastic_grow_worker(rank, world):
    """Resume a single-process (world=1) save at world=2 under ZeRO-1."""
    from luminaai_amd.training import CheckpointManager
    tmp = os.environ["Z1_GROW_TMP"]
    os.chdir(tmp)
    t, cfg = _make_trainer(rank, world, zero_stage=1)
    t.checkpoints = CheckpointManager(os.path.join(tmp, "shared_ckpts"))
    t.load_checkpoint("latest")
    assert t.global_step == 2
    m_sum = float(t.optimizer.groups[0].m.sum())
    master_sum = float(t.optimizer.groups[0].master.sum())
    # training continues after the reshard
    torch.manual_seed(820)
    ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
    t.engine.set_sync(True)
    t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    t.optimizer_step()
    return {"m_sum": m_sum, "master_sum": master_sum,
            "step": t.optimizer.step_count}


def test_zero1_elastic_resume_grow(tmp_path):
    import os as _os
    _os.environ["Z1_GROW_TMP"] = str(tmp_path)
    cwd = _os.getcwd()
    _os.chdir(tmp_path)
    try:
        from luminaai_amd.training import CheckpointManager
        t, cfg = _make_trainer(0, 1, zero_stage=1)
        t.checkpoints = CheckpointManager(str(tmp_path / "shared_ckpts"))
        torch.manual_seed(810)
        for _ in range(2):
            ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
            t.engine.set_sync(True)
            t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
            t.optimizer_step()
        t.save_checkpoint()
        g = t.optimizer.groups[0]
        full_m = float(g.m.sum())
        full_master = float(g.master.sum())
    finally:
        _os.chdir(cwd)
    res = _spawn("zero1_elastic_grow_worker")
    # the two world-2 shards partition the world-1 state exactly
    assert res[0]["m_sum"] + res[1]["m_sum"] == pytest.approx(full_m,
                                                              rel=1e-5)
    assert res[0]["master_sum"] + res[1]["master_sum"] == \
        pytest.approx(full_master, rel=1e-5)
    assert res[0]["step"] == 3


# ---- ring sequence parallelism ---------------------------------------------
def sp_ring_worker(rank, world):
    """sp_mode="ring": forward AND replicated-param grads must match the
    full-sequence single-rank model — including head counts that do NOT
    divide sp (Ulysses' hard limit)."""
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.models.transformer import DeepSeekConfig
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.ops import fused_cross_entropy

    # 3 heads / 1 kv head: indivisible by sp=2 -> only ring can shard this
    mcfg = DeepSeekConfig(vocab_size=512, hidden_size=48, num_layers=2,
                          num_heads=3, num_kv_heads=1, intermediate_size=96,
                          seq_length=64, use_moe=False, use_mod=False,
                          tie_word_embeddings=False)
    reset_mesh()
    torch.manual_seed(1234)
    full = DeepSeekTransformer(mcfg)

    init_mesh(sp_size=world, sp_mode="ring")
    torch.manual_seed(1234)
    sp_model = DeepSeekTransformer(mcfg)
    with torch.no_grad():
        for p_sp, p_f in zip(sp_model.parameters(), full.parameters()):
            p_sp.copy_(p_f)

    torch.manual_seed(777)
    ids = torch.randint(1, mcfg.vocab_size, (2, 65))
    inp, lab = ids[:, :-1], ids[:, 1:]
    S_loc = 64 // world
    lo, hi = rank * S_loc, (rank + 1) * S_loc

    logits_f, _, _ = full(inp)
    ce_f, _, _ = fused_cross_entropy(logits_f, lab)
    ce_f.backward()

    logits_sp, _, _ = sp_model(inp[:, lo:hi])
    torch.testing.assert_close(logits_sp, logits_f[:, lo:hi].detach(),
                               rtol=1e-4, atol=1e-4)
    ce_sp, _, _ = fused_cross_entropy(logits_sp, lab[:, lo:hi])
    # full-model CE averages over ALL tokens; local CE over the shard —
    # rescale so the grad contribution matches, then sum across ranks
    (ce_sp / world).backward()
    g_local = sp_model.embed_tokens.weight.grad.clone()
    dist.all_reduce(g_local)
    err = float((g_local - full.embed_tokens.weight.grad).abs().max()
                / full.embed_tokens.weight.grad.abs().max().clamp_min(1e-12))
    reset_mesh()
    return {"grad_err": err}


def test_sp_ring_matches_full_model():
    res = _spawn("sp_ring_worker")
    for r in range(2):
        assert res[r]["grad_err"] < 1e-3, res


def sync_expert_grads_worker(rank, world):
    """Standalone sync_expert_grads utility: expert grads all-reduce over
    the given replica group, dense grads untouched."""
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.models.transformer import DeepSeekConfig
    from luminaai_amd.parallel.expert_parallel import sync_expert_grads
    torch.manual_seed(1234)
    m = DeepSeekTransformer(DeepSeekConfig(
        vocab_size=128, hidden_size=32, num_layers=1, num_heads=2,
        num_kv_heads=1, intermediate_size=64, seq_length=16, use_moe=True,
        num_experts=2, moe_top_k=1, routing_noise_std=0.0,
        moe_pattern="all", dense_start_layers=0, use_mod=False,
        tie_word_embeddings=False))
    for name, p in m.named_parameters():
        p.grad = torch.full_like(p, float(rank + 1))
    sync_expert_grads(m, dist.group.WORLD)
    gu = m.layers[0].ffn.w_gate_up.grad
    emb = m.embed_tokens.weight.grad
    return {"expert_g": float(gu.flatten()[0]),
            "dense_g": float(emb.flatten()[0])}


def test_sync_expert_grads_utility():
    res = _spawn("sync_expert_grads_worker")
    for r in range(2):
        assert res[r]["expert_g"] == pytest.approx(3.0)  # 1 + 2 summed
        assert res[r]["dense_g"] == pytest.approx(r + 1.0)  # untouched


def sp_ring_trainer_worker(rank, world):
    """Ring SP through the full Trainer (data sharding + optimizer):
    replicas stay in sync after steps."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.training import Trainer
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=64, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=False, use_mod=False,
                 zero_stage=0, precision="fp32",
                 experiment_name=f"spr_t_{rank}",
                 eval_every_n_batches=0, save_every_n_batches=0)
    init_mesh(sp_size=world, sp_mode="ring")
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    torch.manual_seed(911)             # same batch on every SP rank
    for _ in range(2):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
        t.engine.set_sync(True)
        out = t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    w = float(t.model.embed_tokens.weight.detach().sum())
    loss = float(out["ce_loss"])
    reset_mesh()
    return {"w": w, "loss": loss}


def test_sp_ring_through_trainer():
    res = _spawn("sp_ring_trainer_worker")
    assert res[0]["w"] == pytest.approx(res[1]["w"], abs=1e-4)


def _sp_ckpt_body(rank, world, mode):
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.models.transformer import DeepSeekConfig
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.ops import fused_cross_entropy
    mcfg = DeepSeekConfig(vocab_size=512, hidden_size=64, num_layers=2,
                          num_heads=4, num_kv_heads=2, intermediate_size=128,
                          seq_length=64, use_moe=False, use_mod=False,
                          gradient_checkpointing=True,
                          tie_word_embeddings=False)
    reset_mesh()
    torch.manual_seed(1234)
    full = DeepSeekTransformer(mcfg)
    init_mesh(sp_size=world, sp_mode=mode)
    torch.manual_seed(1234)
    m = DeepSeekTransformer(mcfg)
    with torch.no_grad():
        for a, b in zip(m.parameters(), full.parameters()):
            a.copy_(b)
    m.train()
    full.train()
    torch.manual_seed(777)
    ids = torch.randint(1, mcfg.vocab_size, (2, 65))
    inp, lab = ids[:, :-1], ids[:, 1:]
    S_loc = 64 // world
    lo, hi = rank * S_loc, (rank + 1) * S_loc

    lf, _, _ = full(inp)
    cf, _, _ = fused_cross_entropy(lf, lab)
    cf.backward()
    ls, _, _ = m(inp[:, lo:hi])
    cs, _, _ = fused_cross_entropy(ls, lab[:, lo:hi])
    (cs / world).backward()
    g = m.embed_tokens.weight.grad.clone()
    dist.all_reduce(g)
    err = float((g - full.embed_tokens.weight.grad).abs().max()
                / full.embed_tokens.weight.grad.abs().max().clamp_min(1e-12))
    reset_mesh()
    return {"err": err}


def sp_ring_ckpt_worker(rank, world):
    """Ring SP + activation checkpointing: the recompute re-runs the ring
    exchange inside backward — collective order must stay aligned."""
    return _sp_ckpt_body(rank, world, "ring")


def sp_ulysses_ckpt_worker(rank, world):
    """Ulysses SP + activation checkpointing: recompute re-fires the
    head<->sequence all-to-alls in backward."""
    return _sp_ckpt_body(rank, world, "ulysses")


def test_sp_ring_with_checkpointing():
    res = _spawn("sp_ring_ckpt_worker")
    for r in range(2):
        assert res[r]["err"] < 1e-3, res


def test_sp_ulysses_with_checkpointing():
    res = _spawn("sp_ulysses_ckpt_worker")
    for r in range(2):
        assert res[r]["err"] < 1e-3, res


def ep_ckpt_worker(rank, world):
    """EP + activation checkpointing: token all-to-alls re-fire during
    recompute; grads must equal the non-checkpointed EP run."""
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.models.transformer import DeepSeekConfig
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.ops import fused_cross_entropy

    def run(ckpt):
        reset_mesh()
        init_mesh(world)
        mcfg = DeepSeekConfig(vocab_size=512, hidden_size=64, num_layers=2,
                              num_heads=4, num_kv_heads=2,
                              intermediate_size=128, seq_length=32,
                              use_moe=True, num_experts=4, moe_top_k=2,
                              routing_noise_std=0.0, moe_pattern="all",
                              dense_start_layers=0, use_mod=False,
                              gradient_checkpointing=ckpt,
                              tie_word_embeddings=False)
        torch.manual_seed(1234)
        m = DeepSeekTransformer(mcfg)
        m.train()
        torch.manual_seed(930 + rank)
        ids = torch.randint(1, mcfg.vocab_size, (2, 33))
        logits, aux, _ = m(ids[:, :-1])
        ce, _, _ = fused_cross_entropy(logits, ids[:, 1:])
        (ce + aux).backward()
        return (m.embed_tokens.weight.grad.clone(),
                m.layers[0].ffn.w_gate_up.grad.clone())

    g_plain, e_plain = run(False)
    g_ckpt, e_ckpt = run(True)
    return {"g_err": float((g_plain - g_ckpt).abs().max()),
            "e_err": float((e_plain - e_ckpt).abs().max())}


def test_ep_with_checkpointing():
    res = _spawn("ep_ckpt_worker")
    for r in range(2):
        assert res[r]["g_err"] < 1e-5, res
        assert res[r]["e_err"] < 1e-5, res


# ---- expert add/prune under EP (round-2: SURVEY build plan 7.6) ------------
def ep_zero1_worker(rank, world):
    """EP(2) + ZeRO-1 -- the exact default combo bench.py uses multi-GPU
    (zero_stage 1, ep = world). Must train identically to EP + ZeRO-0
    on the same data (sharded-optimizer bookkeeping must not disturb
    the expert comm groups)."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.training import Trainer

    def run(zstage):
        reset_mesh()
        cfg = Config(vocab_size=512, hidden_size=64, num_layers=2,
                     num_heads=4, num_kv_heads=2, seq_length=32,
                     intermediate_size=128, micro_batch_size=2,
                     gradient_accumulation_steps=1, num_workers=0,
                     use_moe=True, num_experts=4, moe_top_k=2,
                     routing_noise_std=0.0, use_mod=False,
                     zero_stage=zstage, precision="fp32",
                     experiment_name=f"epz{zstage}_{rank}",
                     eval_every_n_batches=0, save_every_n_batches=0)
        init_mesh(world)
        torch.manual_seed(1234)
        model = DeepSeekTransformer(config_to_deepseek_config(cfg))
        t = Trainer(model, ConversationTokenizer(), cfg)
        t._setup_scheduler(10)
        torch.manual_seed(640 + rank)
        for _ in range(2):
            ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1))
            t.engine.set_sync(True)
            t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
            t.optimizer_step()
        emb = t.model.embed_tokens.weight.detach().clone()
        gu = t.model.layers[0].ffn.w_gate_up.detach().clone()
        gn = t.optimizer.last_grad_norm()
        t.engine.remove_hooks()
        reset_mesh()
        return emb, gu, gn

    emb1, gu1, gn1 = run(1)
    emb2, gu2, gn2 = run(2)
    emb0, gu0, gn0 = run(0)
    return {"demb1": float((emb1 - emb0).abs().max()),
            "dgu1": float((gu1 - gu0).abs().max()),
            "demb2": float((emb2 - emb0).abs().max()),
            "dgu2": float((gu2 - gu0).abs().max()),
            "gn1": gn1, "gn2": gn2, "gn0": gn0, "rank": rank}


def test_ep_with_zero1_zero2():
    res = _spawn("ep_zero1_worker")
    for r in range(WORLD):
        assert res[r]["demb1"] < 1e-5 and res[r]["dgu1"] < 1e-5, res
        assert res[r]["demb2"] < 1e-5 and res[r]["dgu2"] < 1e-5, res
        assert res[r]["gn1"] == pytest.approx(res[r]["gn0"], rel=1e-5), res
        assert res[r]["gn2"] == pytest.approx(res[r]["gn0"], rel=1e-5), res


def ep_fp8_a2a_worker(rank, world):
    """EP(2) with fp8_alltoall: forward/backward equal the full-expert
    model within e4m3 quantization tolerance (payload e4m3, grads e5m2)."""
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh

    mcfg = _moe_model_cfg()
    reset_mesh()
    torch.manual_seed(1234)
    full = DeepSeekTransformer(mcfg)
    init_mesh(world)
    torch.manual_seed(1234)
    ep = DeepSeekTransformer(mcfg)
    with torch.no_grad():
        for (n, p), (n2, q) in zip(ep.named_parameters(),
                                   full.named_parameters()):
            p.copy_(q if p.shape == q.shape
                    else q[rank * 2:(rank + 1) * 2])
    for l in ep.layers:
        l.ffn.fp8_alltoall = True
    torch.manual_seed(77)
    ids_all = torch.randint(1, mcfg.vocab_size, (2, 2, mcfg.seq_length))
    lf, _, _ = full(ids_all.reshape(4, mcfg.seq_length))
    le, _, _ = ep(ids_all[rank])
    ref = lf.view(2, 2, mcfg.seq_length, -1)[rank]
    rel = (le - ref).abs().max() / ref.abs().max().clamp_min(1e-6)
    # grads flow (e5m2 path) without error
    le.float().pow(2).mean().backward()
    g = ep.layers[0].ffn.w_gate_up.grad
    reset_mesh()
    return {"rel": float(rel), "grad_finite": bool(torch.isfinite(g).all())}


def test_ep_fp8_alltoall():
    res = _spawn("ep_fp8_a2a_worker")
    for r in range(WORLD):
        assert res[r]["rel"] < 0.05, res       # e4m3 rowwise quant band
        assert res[r]["grad_finite"], res


def ep_elastic_save_worker(rank, world):
    """Train EP(2) 2 steps, save, then ONE more step (the reference
    continuation for the elastic single-proc resume)."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.training import CheckpointManager, Trainer
    tmp = os.environ["EP_ELASTIC_TMP"]
    os.chdir(tmp)
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=True, num_experts=4, moe_top_k=2,
                 routing_noise_std=0.0, use_mod=False,
                 zero_stage=0, precision="fp32",
                 experiment_name=f"epel_{rank}",
                 eval_every_n_batches=0, save_every_n_batches=0)
    init_mesh(world)
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    t.checkpoints = CheckpointManager(os.path.join(tmp, "ep_ckpts"))
    gens = [torch.Generator().manual_seed(870 + r) for r in range(world)]
    for _ in range(2):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1),
                            generator=gens[rank])
        t.engine.set_sync(True)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    t.save_checkpoint(tag="elastic_test")
    import torch.distributed as dist
    dist.barrier()
    # reference continuation step (same per-rank shards)
    ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1),
                        generator=gens[rank])
    t.engine.set_sync(True)
    t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    t.optimizer_step()
    emb = float(t.model.embed_tokens.weight.detach().sum())
    gu = t.model.layers[0].ffn.w_gate_up.detach().clone()
    reset_mesh()
    return {"emb": emb, "gu_sum": float(gu.sum()), "rank": rank}


def test_ep_elastic_resume_world1(tmp_path):
    """Elastic EP resume: a world-2/EP-2 run's checkpoint loads into a
    single process (all 4 experts + optimizer moments re-assembled from
    the two _ep_rank files) and continues training identically."""
    os.environ["EP_ELASTIC_TMP"] = str(tmp_path)
    res = _spawn("ep_elastic_save_worker")

    import torch as th
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.training import CheckpointManager, Trainer
    os.environ.pop("WORLD_SIZE", None)
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=4, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=True, num_experts=4, moe_top_k=2,
                 routing_noise_std=0.0, use_mod=False,
                 zero_stage=0, precision="fp32", experiment_name="epel_r",
                 eval_every_n_batches=0, save_every_n_batches=0)
    th.manual_seed(1)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    t.checkpoints = CheckpointManager(str(tmp_path / "ep_ckpts"))
    t.load_checkpoint("latest")
    # continuation on the CONCATENATED global batch of the ref step
    gens = [th.Generator().manual_seed(870 + r) for r in range(2)]
    for g in gens:                      # advance past the 2 pre-save draws
        for _ in range(2):
            th.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1),
                       generator=g)
    rows = [th.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1),
                       generator=g) for g in gens]
    ids = th.cat(rows)
    t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    t.optimizer_step()
    emb = float(t.model.embed_tokens.weight.detach().sum())
    gu = t.model.layers[0].ffn.w_gate_up.detach()     # [4, h, 2I]
    assert emb == pytest.approx(res[0]["emb"], abs=1e-3), \
        (emb, res[0]["emb"])
    for r in range(2):
        ref = float(gu[2 * r:2 * r + 2].sum())
        assert res[r]["gu_sum"] == pytest.approx(ref, abs=1e-3), \
            (r, res[r]["gu_sum"], ref)


def ep_elastic_grow_worker(rank, world):
    """Resume a SINGLE-process MoE checkpoint at EP(2): elastic grow."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.training import CheckpointManager, Trainer
    tmp = os.environ["EP_ELASTIC_TMP"]
    os.chdir(tmp)
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=True, num_experts=4, moe_top_k=2,
                 routing_noise_std=0.0, use_mod=False,
                 zero_stage=0, precision="fp32",
                 experiment_name=f"epgrow_{rank}",
                 eval_every_n_batches=0, save_every_n_batches=0)
    init_mesh(world)
    torch.manual_seed(7)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    t.checkpoints = CheckpointManager(os.path.join(tmp, "grow_ckpts"))
    t.load_checkpoint("latest")
    gens = [torch.Generator().manual_seed(880 + r) for r in range(world)]
    ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1),
                        generator=gens[rank])
    t.engine.set_sync(True)
    t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    t.optimizer_step()
    emb = float(t.model.embed_tokens.weight.detach().sum())
    gu = t.model.layers[0].ffn.w_gate_up.detach().clone()
    reset_mesh()
    return {"emb": emb, "gu_sum": float(gu.sum()), "rank": rank}


def test_ep_elastic_grow_world2(tmp_path):
    """Single-process MoE run saved, resumed under EP(2): expert weights
    AND moments slice across the new ranks; continuation matches the
    single-process continuation."""
    import torch as th
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.training import CheckpointManager, Trainer
    os.environ["EP_ELASTIC_TMP"] = str(tmp_path)
    os.environ.pop("WORLD_SIZE", None)
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=4, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=True, num_experts=4, moe_top_k=2,
                 routing_noise_std=0.0, use_mod=False,
                 zero_stage=0, precision="fp32", experiment_name="epg_r",
                 eval_every_n_batches=0, save_every_n_batches=0)
    th.manual_seed(7)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    t.checkpoints = CheckpointManager(str(tmp_path / "grow_ckpts"))
    gens = [th.Generator().manual_seed(860 + r) for r in range(2)]
    for _ in range(2):
        rows = [th.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1),
                           generator=g) for g in gens]
        ids = th.cat(rows)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    t.save_checkpoint(tag="grow_test")
    # single-process continuation (the reference)
    gens2 = [th.Generator().manual_seed(880 + r) for r in range(2)]
    rows = [th.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1),
                       generator=g) for g in gens2]
    ids = th.cat(rows)
    t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    t.optimizer_step()
    emb_ref = float(t.model.embed_tokens.weight.detach().sum())
    gu = t.model.layers[0].ffn.w_gate_up.detach()

    res = _spawn("ep_elastic_grow_worker")
    for r in range(2):
        assert res[r]["emb"] == pytest.approx(emb_ref, abs=1e-3), \
            (r, res[r]["emb"], emb_ref)
        ref = float(gu[2 * r:2 * r + 2].sum())
        assert res[r]["gu_sum"] == pytest.approx(ref, abs=1e-3), \
            (r, res[r]["gu_sum"], ref)


def tpep_worker(rank, world):
    """TP(2) x EP(2) composed mesh (world 4, dp 1): full Trainer steps.
    Ranks (e, t) hold expert shard e sliced by tp rank t; tp peers share a
    batch shard, ep peers exchange tokens. Must match single-process
    training on the concatenated global batch."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.parallel.tensor_parallel import convert_to_tensor_parallel
    from luminaai_amd.training import Trainer
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=True, num_experts=4, moe_top_k=2,
                 routing_noise_std=0.0, use_mod=False,
                 zero_stage=0, precision="fp32",
                 experiment_name=f"tpep_{rank}",
                 eval_every_n_batches=0, save_every_n_batches=0)
    # build the FULL model first (identical RNG stream to the reference:
    # EP-sharded construction consumes fewer randoms and diverges all
    # later params), then copy dense params + this rank's expert slices
    mcfg = config_to_deepseek_config(cfg)
    torch.manual_seed(1234)
    fullm = DeepSeekTransformer(mcfg)
    for l in fullm.layers:                  # decisive routing: TP float-sum
        l.ffn.gate.weight.data.mul_(50.0)   # reordering must not flip top-k
    mesh = init_mesh(ep_size=2, tp_size=2)
    assert (mesh.dp_size, mesh.ep_rank, mesh.tp_rank) == \
        (1, rank // 2, rank % 2)
    model = DeepSeekTransformer(mcfg)
    EL = 4 // 2
    e0 = mesh.ep_rank * EL
    with torch.no_grad():
        for (n, p), (n2, q) in zip(model.named_parameters(),
                                   fullm.named_parameters()):
            assert n == n2
            p.copy_(q if p.shape == q.shape else q[e0:e0 + EL])
    convert_to_tensor_parallel(model, mesh)
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    drank = rank // 2                       # tp peers share a batch shard
    gens = [torch.Generator().manual_seed(640 + d) for d in range(2)]
    for _ in range(2):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1),
                            generator=gens[drank])
        t.engine.set_sync(True)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    emb = float(t.model.embed_tokens.weight.detach().sum())
    gu = t.model.layers[0].ffn.w_gate_up.detach()
    gn = t.optimizer.last_grad_norm()
    reset_mesh()
    return {"emb": emb, "gu_sum": float(gu.sum()),
            "gu_shape": tuple(gu.shape), "grad_norm": gn, "rank": rank}


def dpep_worker(rank, world):
    """DP(2) x EP(2) (world 4): the expert-replica grad all-reduce
    (expert_dp_group, stride ep) finally has >1 member -- dense params
    sync over the world group, expert shards over their replica pair.
    Must match single-process training on the concatenated global batch."""
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.training import Trainer
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=True, num_experts=4, moe_top_k=2,
                 routing_noise_std=0.0, use_mod=False,
                 zero_stage=0, precision="fp32",
                 experiment_name=f"dpep_{rank}",
                 eval_every_n_batches=0, save_every_n_batches=0)
    mcfg = config_to_deepseek_config(cfg)
    torch.manual_seed(1234)
    fullm = DeepSeekTransformer(mcfg)
    mesh = init_mesh(ep_size=2)
    assert (mesh.dp_size, mesh.ep_rank, mesh.dp_rank) == \
        (2, rank % 2, rank // 2)
    model = DeepSeekTransformer(mcfg)
    e0 = mesh.ep_rank * 2
    with torch.no_grad():
        for (n, p), (n2, q) in zip(model.named_parameters(),
                                   fullm.named_parameters()):
            p.copy_(q if p.shape == q.shape else q[e0:e0 + 2])
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    # data rank = dp_rank (ep peers of one replica share... no: EP ranks
    # have DISTINCT batch shards; 4 ranks = 4 shards)
    gens = [torch.Generator().manual_seed(820 + r) for r in range(4)]
    for _ in range(2):
        ids = torch.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1),
                            generator=gens[rank])
        t.engine.set_sync(True)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    emb = float(t.model.embed_tokens.weight.detach().sum())
    gu = t.model.layers[0].ffn.w_gate_up.detach()
    gn = t.optimizer.last_grad_norm()
    reset_mesh()
    return {"emb": emb, "gu_sum": float(gu.sum()), "grad_norm": gn,
            "rank": rank}


def test_dpep_matches_single_process():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_run, args=(r, 4, port, "dpep_worker", q))
             for r in range(4)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(4):
        rank, status, payload = q.get()
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        res[rank] = payload
    for p in procs:
        p.join(timeout=180)

    import torch as th
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.training import Trainer
    os.environ.pop("WORLD_SIZE", None)
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=8, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=True, num_experts=4, moe_top_k=2,
                 routing_noise_std=0.0, use_mod=False,
                 zero_stage=0, precision="fp32", experiment_name="dpep_ref",
                 eval_every_n_batches=0, save_every_n_batches=0)
    th.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    gens = [th.Generator().manual_seed(820 + r) for r in range(4)]
    for _ in range(2):
        rows = [th.randint(1, cfg.vocab_size, (2, cfg.seq_length + 1),
                           generator=g) for g in gens]
        ids = th.cat(rows)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    emb_ref = float(t.model.embed_tokens.weight.detach().sum())
    gn_ref = t.optimizer.last_grad_norm()
    gu = t.model.layers[0].ffn.w_gate_up.detach()
    for r in range(4):
        e = r % 2
        assert res[r]["emb"] == pytest.approx(emb_ref, abs=1e-3), \
            (r, res[r]["emb"], emb_ref)
        assert res[r]["grad_norm"] / 4 == pytest.approx(gn_ref, rel=2e-3), \
            (r, res[r]["grad_norm"], gn_ref)
        ref = float(gu[2 * e:2 * e + 2].sum())
        assert res[r]["gu_sum"] == pytest.approx(ref, abs=1e-2), \
            (r, res[r]["gu_sum"], ref)


def test_tpep_matches_single_process():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_run, args=(r, 4, port, "tpep_worker", q))
             for r in range(4)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(4):
        rank, status, payload = q.get()
        assert status == "ok", f"rank {rank} failed:\n{payload}"
        res[rank] = payload
    for p in procs:
        p.join(timeout=180)

    # single-process reference on the concatenated global batch
    import torch as th
    from luminaai_amd.config import Config
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.training import Trainer
    os.environ.pop("WORLD_SIZE", None)
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=4, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=True, num_experts=4, moe_top_k=2,
                 routing_noise_std=0.0, use_mod=False,
                 zero_stage=0, precision="fp32", experiment_name="tpep_ref",
                 eval_every_n_batches=0, save_every_n_batches=0)
    th.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    for l in model.layers:
        l.ffn.gate.weight.data.mul_(50.0)
    t = Trainer(model, ConversationTokenizer(), cfg)
    t._setup_scheduler(10)
    gens = [th.Generator().manual_seed(640 + d) for d in range(2)]
    for _ in range(2):
        rows = []
        for g in gens:
            rows.append(th.randint(1, cfg.vocab_size,
                                   (2, cfg.seq_length + 1), generator=g))
        ids = th.cat(rows)
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
    emb_ref = float(t.model.embed_tokens.weight.detach().sum())
    gn_ref = t.optimizer.last_grad_norm()
    gu = t.model.layers[0].ffn.w_gate_up.detach()      # [4, h, 2I]
    I = cfg.intermediate_size
    li = I // 2
    for r in range(4):
        e, tp = r // 2, r % 2
        assert res[r]["emb"] == pytest.approx(emb_ref, abs=1e-3), \
            (r, res[r]["emb"], emb_ref)
        # the GLOBAL grad norm is the cross-class scaling check (sums the
        # dense/"tp"/"expert" comm classes over their exchange groups);
        # Adam updates are scale-invariant, weight sums alone can't see it.
        # last_grad_norm is the RAW (pre-grad_scale) norm: distributed
        # grads are world-summed, so raw norm = world x single-proc norm
        assert res[r]["grad_norm"] / 4 == pytest.approx(gn_ref, rel=2e-3), \
            (r, res[r]["grad_norm"], gn_ref)
        ref = th.cat([gu[2 * e:2 * e + 2, :, tp * li:(tp + 1) * li],
                      gu[2 * e:2 * e + 2, :, I + tp * li:I + (tp + 1) * li]],
                     dim=2)
        assert res[r]["gu_shape"] == tuple(ref.shape), \
            (res[r]["gu_shape"], tuple(ref.shape))
        # expert weights drift by Adam-amplified low-bit TP reorder noise
        assert res[r]["gu_sum"] == pytest.approx(float(ref.sum()),
                                                 abs=2e-2), \
            (r, res[r]["gu_sum"], float(ref.sum()))


def ep_add_prune_worker(rank, world):
    """add_expert/prune_expert under ep=2: shards stay even, every rank ends
    with an identical full expert stack, and the resharded model's forward
    equals a fresh full-expert (ep=1) model built from the gathered stack."""
    import torch.distributed as dist
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh

    mcfg = _moe_model_cfg()
    init_mesh(world)
    torch.manual_seed(1234 + rank)   # rank-dependent on purpose: the seed
    model = DeepSeekTransformer(mcfg)  # broadcast must still align shards
    with torch.no_grad():             # re-sync replicated params from rank 0
        for p in model.parameters():
            dist.broadcast(p.data, src=0)
    ffn = model.layers[0].ffn
    E0 = ffn.num_experts

    for l in model.get_moe_layers():
        l.add_expert()
    assert ffn.num_experts == E0 + world
    assert ffn.w_gate_up.shape[0] == (E0 + world) // world
    assert ffn.gate.weight.shape[0] == E0 + world

    # all ranks hold consistent shards: gathered stack equals rank 0's
    full_gu, full_dn = ffn._gather_full_experts()
    ref_gu = full_gu.clone()
    dist.broadcast(ref_gu, src=0)
    torch.testing.assert_close(full_gu, ref_gu)

    # resharded forward == single-process full-expert model on same weights
    reset_mesh()
    from luminaai_amd.models.transformer import DeepSeekConfig
    mcfg2 = DeepSeekConfig(**{**mcfg.__dict__, "num_experts": E0 + world})
    torch.manual_seed(77)
    ref = DeepSeekTransformer(mcfg2)
    with torch.no_grad():
        rp = dict(ref.named_parameters())
        mp = dict(model.named_parameters())
        for name, p in rp.items():
            if ".w_gate_up" in name or ".w_down" in name:
                lid = int(name.split(".")[1])
                fgu, fdn = model.layers[lid].ffn._gather_full_experts()
                p.copy_(fgu if ".w_gate_up" in name else fdn)
            else:
                p.copy_(mp[name])
    torch.manual_seed(55)
    ids = torch.randint(1, mcfg.vocab_size, (2, mcfg.seq_length))
    model.eval(); ref.eval()
    # model still has ep mesh groups captured in the layer; forward uses them
    lt, _, _ = model(ids)
    lr, _, _ = ref(ids)
    torch.testing.assert_close(lt, lr, rtol=1e-4, atol=1e-4)

    # prune back down: world least-used experts dropped evenly
    from luminaai_amd.parallel.mesh import init_mesh as _im
    for l in model.get_moe_layers():
        l.prune_expert()
    assert ffn.num_experts == E0
    assert ffn.w_gate_up.shape[0] == E0 // world
    return {"ok": True}


def test_ep_add_prune_expert():
    res = _spawn("ep_add_prune_worker")
    assert res[0]["ok"] and res[1]["ok"]


# ---- TP over MoE layers (round-2: VERDICT item 10) -------------------------
def tp_moe_forward_worker(rank, world):
    """TP(2) over the batched expert weights: forward equals the full model,
    expert-shard grads equal the full-model grad slices."""
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.parallel.mesh import init_mesh, reset_mesh
    from luminaai_amd.parallel.tensor_parallel import convert_to_tensor_parallel

    mcfg = _moe_model_cfg()
    reset_mesh()
    torch.manual_seed(1234)
    full = DeepSeekTransformer(mcfg)

    mesh = init_mesh(tp_size=world)
    torch.manual_seed(1234)
    tpm = DeepSeekTransformer(mcfg)
    with torch.no_grad():
        for a, b in zip(tpm.parameters(), full.parameters()):
            a.copy_(b)
    n = convert_to_tensor_parallel(tpm, mesh)
    assert n == 2
    li = mcfg.intermediate_size // world
    assert tpm.layers[0].ffn.w_gate_up.shape[2] == 2 * li
    assert tpm.layers[0].ffn.w_down.shape[1] == li

    torch.manual_seed(777)
    ids = torch.randint(1, mcfg.vocab_size, (2, mcfg.seq_length))
    lf, auxf, _ = full(ids)
    lt, auxt, _ = tpm(ids)
    torch.testing.assert_close(lt, lf, rtol=2e-4, atol=2e-4)
    torch.testing.assert_close(auxt, auxf, rtol=1e-4, atol=1e-5)

    lf.float().pow(2).mean().backward()
    lt.float().pow(2).mean().backward()
    # sharded expert w_down: TP grad == full grad slice on the I dim
    gt = tpm.layers[0].ffn.w_down.grad
    gf = full.layers[0].ffn.w_down.grad[:, rank * li:(rank + 1) * li, :]
    torch.testing.assert_close(gt, gf, rtol=0.05,
                               atol=1e-4 * max(gf.abs().max().item(), 1e-3))
    reset_mesh()
    return {"ok": True}


def test_tp_moe_forward_backward():
    res = _spawn("tp_moe_forward_worker")
    assert res[0]["ok"] and res[1]["ok"]
