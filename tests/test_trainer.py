"""Trainer tests: loss, steps, adaptive API, checkpointing, schedulers."""

import math
import os

import pytest
import torch

from luminaai_amd.data.dataset import SyntheticDataset, create_dataloader
from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
from luminaai_amd.training import Trainer, WarmupScheduler


@pytest.fixture
def trainer(tiny_moe_config, tokenizer, tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    torch.manual_seed(0)
    model = DeepSeekTransformer(config_to_deepseek_config(tiny_moe_config))
    return Trainer(model, tokenizer, tiny_moe_config)


def _batch(cfg, bs=2):
    ids = torch.randint(1, cfg.vocab_size, (bs, cfg.seq_length + 1))
    return {"input_ids": ids[:, :-1], "labels": ids[:, 1:],
            "loss_weights": torch.ones(bs, cfg.seq_length)}


def test_train_step_and_optimizer_step(trainer, tiny_moe_config):
    p0 = trainer.model.embed_tokens.weight.detach().clone()
    out = trainer.train_step(_batch(tiny_moe_config))
    assert float(out["loss"]) > 0
    trainer._setup_scheduler(10)
    trainer.optimizer_step()
    assert trainer.global_step == 1
    assert not torch.allclose(trainer.model.embed_tokens.weight.detach(), p0)


def test_loss_decreases(trainer, tiny_moe_config):
    trainer._setup_scheduler(30)
    batch = _batch(tiny_moe_config, bs=2)
    losses = []
    for _ in range(15):
        out = trainer.train_step(batch)
        trainer.optimizer_step()
        losses.append(float(out["ce_loss"].detach()))
    assert losses[-1] < losses[0]  # memorizes a fixed batch


def test_grad_accumulation_equivalence(tiny_config, tokenizer, tmp_path, monkeypatch):
    """Two micro-batches with accum=2 == one combined batch (same grads)."""
    monkeypatch.chdir(tmp_path)
    torch.manual_seed(0)
    b1 = _batch(tiny_config, bs=2)
    b2 = _batch(tiny_config, bs=2)
    combined = {k: torch.cat([b1[k], b2[k]]) for k in b1}

    def fresh():
        torch.manual_seed(42)
        model = DeepSeekTransformer(config_to_deepseek_config(tiny_config))
        return Trainer(model, tokenizer, tiny_config)

    ta = fresh()
    ta.train_step(b1)
    ta.train_step(b2)
    ga = ta.optimizer.groups[0].flat_g.clone() / 2  # summed over 2 micro

    tb = fresh()
    tb.train_step(combined)
    gb = tb.optimizer.groups[0].flat_g.clone()
    # combined batch averages over 2x tokens; accumulation sums two averages
    assert torch.allclose(ga, gb / 1.0, atol=2e-3)


def test_evaluate_reports_mean(trainer, tiny_moe_config):
    ds = SyntheticDataset(tiny_moe_config.vocab_size, tiny_moe_config.seq_length, 8)
    dl = create_dataloader(ds, tiny_moe_config, shuffle=False)
    res = trainer.evaluate(dl)
    assert "loss" in res and "perplexity" in res and res["batches"] == 4
    assert res["perplexity"] == pytest.approx(math.exp(min(res["loss"], 20)), rel=1e-4)


def test_adjust_learning_rate(trainer):
    trainer._setup_scheduler(100)
    trainer.config.min_override_threshold = 0.0
    ok = trainer.adjust_learning_rate(1e-3, grace_period=5)
    assert ok
    assert trainer.get_lr() == pytest.approx(1e-3, rel=0.2)


def test_emergency_lr(trainer):
    trainer._setup_scheduler(100)
    lr0 = trainer.get_lr()
    new = trainer.emergency_lr_reduction(0.1)
    assert new == pytest.approx(lr0 * 0.1)


def test_add_prune_expert_through_trainer(trainer, tiny_moe_config):
    n0 = trainer._moe_layers()[0].num_experts
    assert trainer.add_expert()
    assert trainer._moe_layers()[0].num_experts == n0 + 1
    # training still works after rebuild
    out = trainer.train_step(_batch(tiny_moe_config))
    trainer.optimizer_step()
    assert torch.isfinite(out["loss"]).all()
    assert trainer.prune_expert()
    assert trainer._moe_layers()[0].num_experts == n0
    out = trainer.train_step(_batch(tiny_moe_config))
    trainer.optimizer_step()
    assert torch.isfinite(out["loss"]).all()


def test_moe_stats_and_interventions(trainer):
    trainer.train_step(_batch(trainer.config))
    stats = trainer.get_expert_statistics()
    assert stats["summary"]["num_experts"] == 4
    trainer.adjust_capacity_factor(2.0)
    assert trainer._moe_layers()[0].capacity_factor == 2.0
    trainer.adjust_routing_temperature(0.5)
    trainer.enable_expert_dropout(0.1)
    assert trainer._moe_layers()[0].expert_dropout == 0.1


def test_checkpoint_save_load_roundtrip(trainer, tiny_moe_config):
    trainer._setup_scheduler(10)
    trainer.train_step(_batch(tiny_moe_config))
    trainer.optimizer_step()
    path = trainer.save_checkpoint()
    assert os.path.exists(path)
    w0 = trainer.model.embed_tokens.weight.detach().clone()
    step0 = trainer.global_step
    # perturb then restore
    with torch.no_grad():
        trainer.model.embed_tokens.weight.add_(1.0)
    trainer.load_checkpoint("latest")
    assert torch.allclose(trainer.model.embed_tokens.weight.detach(), w0)
    assert trainer.global_step == step0


def test_checkpoint_resume_equivalence(tiny_config, tokenizer, tmp_path, monkeypatch):
    """Training N steps == training k, save, load, train N-k (same weights)."""
    monkeypatch.chdir(tmp_path)
    batches = [_batch(tiny_config) for _ in range(4)]

    def fresh():
        torch.manual_seed(7)
        model = DeepSeekTransformer(config_to_deepseek_config(tiny_config))
        t = Trainer(model, tokenizer, tiny_config)
        t._setup_scheduler(10)
        return t

    ta = fresh()
    for b in batches:
        ta.train_step(b)
        ta.optimizer_step()
    wa = ta.model.embed_tokens.weight.detach().clone()

    tb = fresh()
    for b in batches[:2]:
        tb.train_step(b)
        tb.optimizer_step()
    tb.save_checkpoint(tag="mid")
    tc = fresh()
    tc.load_checkpoint("latest")
    for b in batches[2:]:
        tc.train_step(b)
        tc.optimizer_step()
    wc = tc.model.embed_tokens.weight.detach().clone()
    assert torch.allclose(wa, wc, atol=1e-5)


def test_rollback_steps(trainer, tiny_moe_config):
    trainer._setup_scheduler(20)
    for _ in range(3):
        trainer.train_step(_batch(tiny_moe_config))
        trainer.optimizer_step()
        trainer.save_checkpoint()
    step3 = trainer.global_step
    for _ in range(2):
        trainer.train_step(_batch(tiny_moe_config))
        trainer.optimizer_step()
    assert trainer.rollback_steps(2)
    assert trainer.global_step <= step3


def test_mod_interventions(tiny_config, tokenizer, tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    tiny_config.use_mod = True
    tiny_config.mod_capacity_factor = 0.5
    model = DeepSeekTransformer(config_to_deepseek_config(tiny_config))
    t = Trainer(model, tokenizer, tiny_config)
    t.train_step(_batch(tiny_config))
    stats = t.get_mod_statistics()
    assert stats["mean_skip_frac"] == pytest.approx(0.5, abs=0.1)
    t.adjust_mod_capacity(0.8)
    assert t.model.layers[0].mod_router.capacity_factor == 0.8


def test_scheduler_shapes():
    class FakeOpt:
        def __init__(self):
            class G:
                lr = 1e-3
            self.groups = [G()]
    opt = FakeOpt()
    s = WarmupScheduler(opt, total_steps=100, warmup_steps=10, kind="cosine",
                        min_lr=1e-6)
    lrs = []
    for _ in range(100):
        s.step()
        lrs.append(opt.groups[0].lr)
    assert lrs[8] > lrs[0]          # warmup rises
    assert lrs[-1] < lrs[20]        # cosine decays
    assert min(lrs) >= 1e-6


def test_nan_batch_does_not_poison_weights(trainer, tiny_moe_config):
    trainer._setup_scheduler(10)
    batch = _batch(tiny_moe_config)
    trainer.train_step(batch)
    trainer.optimizer_step()
    w0 = trainer.model.embed_tokens.weight.detach().clone()
    # poison grads directly, then step: NaN-skip must leave weights unchanged
    trainer.optimizer.groups[0].flat_g.fill_(float("nan"))
    trainer.engine.step(grad_scale=1.0)
    assert torch.isfinite(trainer.model.embed_tokens.weight.detach()).all()
    assert torch.allclose(trainer.model.embed_tokens.weight.detach(), w0)


def test_loss_decreases_on_learnable_data(tiny_config, tokenizer):
    """Loss-curve regression: a tiny model on repetitive data must learn
    (the reference had no such test — SURVEY.md §4 gap list)."""
    import torch
    from luminaai_amd.data.dataset import ConversationDataset, create_dataloader
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.training import Trainer
    from luminaai_amd.utils import generate_sample_data
    import tempfile, os
    with tempfile.TemporaryDirectory() as d:
        p = os.path.join(d, "train.jsonl")
        generate_sample_data(p, n=16, seed=0)
        cfg = tiny_config
        cfg.learning_rate = 2e-3
        cfg.num_epochs = 4
        torch.manual_seed(0)
        model = DeepSeekTransformer(config_to_deepseek_config(cfg))
        t = Trainer(model, tokenizer, cfg)
        ds = ConversationDataset(p, tokenizer, cfg.seq_length)
        dl = create_dataloader(ds, cfg, shuffle=False)
        t._setup_scheduler(64)
        first = None
        for epoch in range(4):
            stats = t.train_epoch(dl, epoch)
            if first is None:
                first = stats["mean_loss"]
        assert stats["mean_loss"] < first * 0.8, \
            f"no learning: {first} -> {stats['mean_loss']}"


def test_profile_training_loop_overhead(tiny_config, tokenizer, small_model):
    from luminaai_amd.training import Trainer
    import torch
    t = Trainer(small_model, tokenizer, tiny_config)
    t._setup_scheduler(10)
    ids = torch.randint(1, 512, (2, 33))
    res = t.profile_training_loop_overhead(
        {"input_ids": ids[:, :-1], "labels": ids[:, 1:]}, iters=2)
    assert set(res) == {"h2d_ms", "forward_ms", "backward_ms", "optimizer_ms"}
    assert all(v >= 0 for v in res.values())


def test_scheduler_kinds():
    import torch
    from luminaai_amd.training.optimizer import FlatAdamW
    from luminaai_amd.training.schedulers import WarmupScheduler
    m = torch.nn.Linear(8, 8)
    for kind in ("cosine", "linear", "constant"):
        opt = FlatAdamW(m, lr=1e-3)
        s = WarmupScheduler(opt, total_steps=100, warmup_steps=10, kind=kind,
                            min_lr=1e-6)
        lrs = []
        for _ in range(100):
            s.step()
            lrs.append(opt.groups[0].lr)
        assert lrs[5] < lrs[7] <= 1e-3            # warmup ramps
        if kind == "constant":
            assert lrs[-1] == pytest.approx(1e-3)
        else:
            assert lrs[-1] < lrs[20]              # decays
            assert lrs[-1] >= 1e-6
        # adaptive override rebasing
        s.set_base_lr(5e-4)
        s.step()
        assert opt.groups[0].lr <= 5e-4 + 1e-9


def test_scheduler_state_roundtrip():
    import torch
    from luminaai_amd.training.optimizer import FlatAdamW
    from luminaai_amd.training.schedulers import WarmupScheduler
    m = torch.nn.Linear(8, 8)
    opt = FlatAdamW(m, lr=1e-3)
    s = WarmupScheduler(opt, 100, 10)
    for _ in range(37):
        s.step()
    sd = s.state_dict()
    opt2 = FlatAdamW(torch.nn.Linear(8, 8), lr=1e-3)
    s2 = WarmupScheduler(opt2, 100, 10)
    s2.load_state_dict(sd)
    assert s2.step_num == 37
    assert opt2.groups[0].lr == pytest.approx(opt.groups[0].lr)


def test_fp16_loss_scaling(tokenizer):
    """fp16 trains with dynamic loss scaling; overflow halves the scale."""
    import torch
    from luminaai_amd.config import Config
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.training import Trainer
    cfg = Config(vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
                 num_kv_heads=2, seq_length=32, intermediate_size=128,
                 micro_batch_size=2, gradient_accumulation_steps=1,
                 num_workers=0, use_moe=False, use_mod=False, zero_stage=0,
                 precision="fp16", experiment_name="fp16_test",
                 eval_every_n_batches=0, save_every_n_batches=0)
    torch.manual_seed(0)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, tokenizer, cfg)
    t._setup_scheduler(10)
    assert t.loss_scale == 65536.0
    w0 = t.model.embed_tokens.weight.detach().float().clone()
    ids = torch.randint(1, 512, (2, 33))
    t.engine.set_sync(True)
    out = t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
    t.optimizer_step()
    assert torch.isfinite(out["ce_loss"].detach().float())
    assert not torch.allclose(t.model.embed_tokens.weight.detach().float(), w0)
    # simulated overflow -> scale backs off
    t.optimizer._last_norm_sq = torch.tensor([float("inf")])
    t._adjust_loss_scale()
    assert t.loss_scale == 32768.0


def test_early_stopping_via_eval(tiny_config, tokenizer, small_model):
    import torch
    from torch.utils.data import DataLoader, TensorDataset
    from luminaai_amd.training import Trainer
    tiny_config.early_stopping_patience = 1
    t = Trainer(small_model, tokenizer, tiny_config)
    t._setup_scheduler(10)
    torch.manual_seed(0)
    ids = torch.randint(1, tiny_config.vocab_size, (4, tiny_config.seq_length + 1))
    eval_batches = [{"input_ids": ids[:, :-1], "labels": ids[:, 1:]}]
    r1 = t.evaluate(eval_batches)
    assert not t.should_stop
    # train a bit with high LR so eval on the SAME random data worsens or at
    # best stays: force the pathway by faking a best
    t.best_eval_loss = -1.0   # anything is "no improvement" now
    t._no_improve_evals = 0
    t.evaluate(eval_batches)
    assert t.should_stop, "patience=1 with no improvement must stop"


def test_training_continues_after_expert_add(tiny_moe_config, tokenizer):
    import torch
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.training import Trainer
    torch.manual_seed(0)
    model = DeepSeekTransformer(config_to_deepseek_config(tiny_moe_config))
    t = Trainer(model, tokenizer, tiny_moe_config)
    t._setup_scheduler(20)

    def step():
        ids = torch.randint(1, tiny_moe_config.vocab_size,
                            (2, tiny_moe_config.seq_length + 1))
        t.engine.set_sync(True)
        out = t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
        return float(out["ce_loss"].detach())

    step()
    e0 = t.model.get_moe_layers()[0].num_experts
    assert t.add_expert()
    assert t.model.get_moe_layers()[0].num_experts == e0 + 1
    loss = step()                      # optimizer rebuilt; training continues
    assert loss == loss and loss > 0
    assert t.prune_expert()
    assert t.model.get_moe_layers()[0].num_experts == e0
    loss = step()
    assert loss == loss and loss > 0


def test_elastic_reshard_world2_to_world1():
    """ZeRO-1/2 checkpoints saved at world=2 resume exactly at world=1 via
    FlatAdamW.load_resharded (shards concatenate back to the flat buffer)."""
    import torch.nn as nn
    from luminaai_amd.training.optimizer import FlatAdamW

    def build():
        torch.manual_seed(42)
        return nn.Sequential(nn.Linear(32, 64), nn.SiLU(),
                             nn.Linear(64, 32)).to(torch.bfloat16)

    def grads_for(step):
        torch.manual_seed(100 + step)
        return None  # grads produced by identical data below

    # reference: world-1 training
    m1 = build()
    o1 = FlatAdamW(m1, lr=1e-2, weight_decay=0.01)
    # simulated world-2 "ranks": two replicas with identical data
    m2a, m2b = build(), build()
    o2a = FlatAdamW(m2a, lr=1e-2, weight_decay=0.01,
                    shard_rank=0, shard_world=2)
    o2b = FlatAdamW(m2b, lr=1e-2, weight_decay=0.01,
                    shard_rank=1, shard_world=2)
    for step in range(3):
        torch.manual_seed(100 + step)
        x = torch.randn(4, 32).to(torch.bfloat16)
        for m, o in ((m1, o1), (m2a, o2a), (m2b, o2b)):
            o.zero_grad()
            m(x).float().pow(2).mean().backward()
            o.step()
        # ZeRO-1 weight all-gather: each rank takes the other's shard
        for ga, gb in zip(o2a.groups, o2b.groups):
            ga.flat_w[gb.shard_lo:gb.shard_hi].copy_(
                gb.flat_w[gb.shard_lo:gb.shard_hi])
            gb.flat_w[ga.shard_lo:ga.shard_hi].copy_(
                ga.flat_w[ga.shard_lo:ga.shard_hi])
    # sanity: the sharded pair tracked the single-rank run
    for p1, p2 in zip(m1.parameters(), m2a.parameters()):
        torch.testing.assert_close(p1, p2)

    # elastic resume at world=1 from the two world-2 shards
    m_new = build()
    o_new = FlatAdamW(m_new, lr=1e-2, weight_decay=0.01)
    with pytest.raises(ValueError, match="load_resharded"):
        o_new.load_state_dict(o2a.state_dict())
    m_new.load_state_dict(m2a.state_dict())
    o_new.load_resharded([o2a.state_dict(), o2b.state_dict()])
    assert o_new.step_count == o1.step_count
    for gn, g1 in zip(o_new.groups, o1.groups):
        n = gn.numel
        torch.testing.assert_close(gn.master[:n], g1.master[:n])
        torch.testing.assert_close(gn.m[:n], g1.m[:n])
        torch.testing.assert_close(gn.v[:n], g1.v[:n])
    # one more identical step stays in lockstep with the reference
    torch.manual_seed(200)
    x = torch.randn(4, 32).to(torch.bfloat16)
    for m, o in ((m1, o1), (m_new, o_new)):
        o.zero_grad()
        m(x).float().pow(2).mean().backward()
        o.step()
    for p1, p2 in zip(m1.parameters(), m_new.parameters()):
        torch.testing.assert_close(p1, p2)


def test_profile_communication_flag(tiny_config, tokenizer):
    """profile_communication populates the decorator-profiling stats with
    the ZeroEngine sync points."""
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.training import Trainer
    from luminaai_amd.utils.profiling import (enable_profiling,
                                              get_profiling_stats,
                                              reset_profiling_stats)
    reset_profiling_stats()
    tiny_config.profile_communication = True
    torch.manual_seed(0)
    model = DeepSeekTransformer(config_to_deepseek_config(tiny_config))
    t = Trainer(model, tokenizer, tiny_config)
    try:
        ids = torch.randint(1, tiny_config.vocab_size,
                            (2, tiny_config.seq_length + 1))
        t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
        stats = get_profiling_stats()
        assert any(k.startswith("comm.") for k in stats), stats.keys()
    finally:
        enable_profiling(False)
        reset_profiling_stats()


def test_adjust_batch_size_and_weight_decay(tiny_config, tokenizer):
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.training import Trainer
    torch.manual_seed(0)
    t = Trainer(DeepSeekTransformer(config_to_deepseek_config(tiny_config)),
                tokenizer, tiny_config)
    t.adjust_batch_size(4)
    assert t.config.micro_batch_size == 4
    t.adjust_batch_size(0)       # clamps to 1
    assert t.config.micro_batch_size == 1
    t.adjust_weight_decay(0.05)
    decayed = [g for g in t.optimizer.groups if g.weight_decay > 0]
    assert decayed and all(g.weight_decay == 0.05 for g in decayed)
    zero_wd = [g for g in t.optimizer.groups if g.weight_decay == 0.0]
    assert zero_wd  # norms/biases stay decay-free
    from luminaai_amd.data.dataset import SyntheticDataset
    dl = t._recreate_dataloader(SyntheticDataset(tiny_config.vocab_size,
                                                 tiny_config.seq_length, 8,
                                                 seed=0))
    assert dl.batch_size == 1


def test_train_with_oom_fallback(tiny_config, tokenizer, monkeypatch):
    """OOM in train() halves micro-batch / doubles accumulation and
    retries (reference Main.py:292-501)."""
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.training import Trainer
    torch.manual_seed(0)
    tiny_config.micro_batch_size = 4
    t = Trainer(DeepSeekTransformer(config_to_deepseek_config(tiny_config)),
                tokenizer, tiny_config)
    calls = {"n": 0}

    def fake_train(*a, **kw):
        calls["n"] += 1
        if calls["n"] < 3:
            raise torch.cuda.OutOfMemoryError("synthetic OOM")
        return {"global_step": 1}

    monkeypatch.setattr(t, "train", fake_train)
    monkeypatch.setattr(torch.cuda, "empty_cache", lambda: None)
    out = t.train_with_oom_fallback()
    assert out["global_step"] == 1
    assert calls["n"] == 3
    assert t.config.micro_batch_size == 1        # 4 -> 2 -> 1


def test_loss_curve_regression(tiny_config, tokenizer):
    """Fixed-seed 30-step run must actually LEARN (loss drops >30% and
    lands under an absolute bound) — catches silent training-math
    regressions (a gap in the reference's own suite)."""
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.training import Trainer
    tiny_config.learning_rate = 3e-3   # tiny model memorising 4 batches
    torch.manual_seed(1234)
    model = DeepSeekTransformer(config_to_deepseek_config(tiny_config))
    t = Trainer(model, tokenizer, tiny_config)
    t._setup_scheduler(80)
    torch.manual_seed(4321)
    # a SMALL fixed corpus, repeated: the model must memorise it
    data = [torch.randint(1, tiny_config.vocab_size,
                          (2, tiny_config.seq_length + 1)) for _ in range(4)]
    losses = []
    for step in range(60):
        ids = data[step % len(data)]
        t.engine.set_sync(True)
        out = t.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        t.optimizer_step()
        losses.append(float(out["ce_loss"]))
    first = sum(losses[:4]) / 4
    last = sum(losses[-4:]) / 4
    assert last < first * 0.7, (first, last)
    assert last < 5.0, losses[-4:]


def test_training_does_not_leak_objects(tiny_config, tokenizer):
    """Steady-state steps must not grow live Python objects (reference
    test_performance.py:108-121 gc leak check)."""
    import gc
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.training import Trainer
    torch.manual_seed(0)
    model = DeepSeekTransformer(config_to_deepseek_config(tiny_config))
    t = Trainer(model, tokenizer, tiny_config)
    ids = torch.randint(1, tiny_config.vocab_size,
                        (2, tiny_config.seq_length + 1))
    batch = {"input_ids": ids[:, :-1], "labels": ids[:, 1:]}

    def steps(n):
        for _ in range(n):
            t.engine.set_sync(True)
            t.train_step(batch)
            t.optimizer_step()

    steps(5)                 # warm the caches
    gc.collect()
    before = len(gc.get_objects())
    steps(10)
    gc.collect()
    after = len(gc.get_objects())
    assert after - before < 500, (before, after)


def test_precision_manager_registry():
    from luminaai_amd.training.precision import PrecisionManager

    class C:
        precision = "bf16"

    pm = PrecisionManager(C())
    assert pm.param_dtype == torch.bfloat16
    assert not pm.needs_loss_scale
    C.precision = "fp16"
    assert PrecisionManager(C()).needs_loss_scale
    C.precision = "auto"
    assert PrecisionManager(C()).spec.name in ("fp32", "bf16")
    C.precision = "int8"         # reference-compat alias -> bf16 training
    assert PrecisionManager(C()).param_dtype == torch.bfloat16
    C.precision = "nope"
    import pytest as _pytest
    with _pytest.raises(ValueError):
        PrecisionManager(C())
    assert "fp8" in PrecisionManager.available()
    # cast_model applies the param dtype
    import torch.nn as nn
    C.precision = "fp16"
    m = PrecisionManager(C()).cast_model(nn.Linear(4, 4))
    assert m.weight.dtype == torch.float16
