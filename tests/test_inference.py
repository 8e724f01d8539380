"""Inference tests: KV-cache correctness, sampling, smart loaders, chat."""

import math

import pytest
import torch

from luminaai_amd.inference import (ChatInterface, GenerationConfig,
                                    GenerationEngine,
                                    infer_config_from_state_dict,
                                    load_checkpoint_smart)
from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config


def test_kv_cache_matches_full_forward(small_model):
    """Incremental decode with KV caches must reproduce the full-sequence
    logits (the reference never exercised its cache path, Chat.py:381)."""
    m = small_model.eval()
    ids = torch.randint(0, 500, (1, 12))
    with torch.no_grad():
        full, _, _ = m(ids)
        caches = m.make_kv_caches()
        pre, _, _ = m(ids[:, :8], kv_caches=caches)
        outs = [pre[:, -1]]
        for t in range(8, 12):
            step, _, _ = m(ids[:, t:t + 1], kv_caches=caches)
            outs.append(step[:, -1])
    inc = torch.stack(outs, dim=1)
    torch.testing.assert_close(inc, full[:, 7:], rtol=1e-4, atol=1e-4)


def test_kv_cache_moe_model(tiny_moe_config, tokenizer):
    torch.manual_seed(0)
    m = DeepSeekTransformer(config_to_deepseek_config(tiny_moe_config)).eval()
    eng = GenerationEngine(m, tokenizer)
    out = eng.generate(tokenizer.encode("hello world"),
                       GenerationConfig(max_new_tokens=8, temperature=0.0))
    assert len(out) <= 8
    assert all(isinstance(t, int) for t in out)
    assert eng.get_stats()["prefill_tokens"] > 0


def test_greedy_deterministic(small_model, tokenizer):
    eng = GenerationEngine(small_model.eval(), tokenizer)
    cfg = GenerationConfig(max_new_tokens=6, temperature=0.0)
    a = eng.generate(tokenizer.encode("abc"), cfg)
    b = eng.generate(tokenizer.encode("abc"), cfg)
    assert a == b


def test_sampling_top_k_top_p(small_model, tokenizer):
    eng = GenerationEngine(small_model.eval(), tokenizer)
    logits = torch.zeros(512)
    logits[7] = 10.0   # dominant token
    cfg = GenerationConfig(temperature=1.0, top_k=1, top_p=1.0,
                           repetition_penalty=1.0)
    assert eng._sample(logits.clone(), cfg, []) == 7
    cfg = GenerationConfig(temperature=1.0, top_k=0, top_p=0.01,
                           repetition_penalty=1.0)
    assert eng._sample(logits.clone(), cfg, []) == 7


def test_repetition_penalty_discourages(small_model, tokenizer):
    eng = GenerationEngine(small_model.eval(), tokenizer)
    logits = torch.zeros(512)
    logits[3], logits[4] = 5.0, 4.9
    cfg = GenerationConfig(temperature=0.0, repetition_penalty=2.0)
    # token 3 recently used -> its logit halves -> 4 wins
    assert eng._sample(logits.clone(), cfg, [3]) == 4


def test_generation_modes():
    g = GenerationConfig.from_mode("creative")
    assert g.temperature == pytest.approx(1.1)
    g = GenerationConfig.from_mode("greedy", max_new_tokens=3)
    assert g.temperature == 0.0 and g.max_new_tokens == 3


# ---------------------------------------------------------------- loaders
def test_load_checkpoint_smart_strips_prefixes(small_model, tmp_path):
    sd = {"module." + k: v for k, v in small_model.state_dict().items()}
    p = str(tmp_path / "ck.pt")
    torch.save({"model_state_dict": sd, "global_step": 7}, p)
    payload = load_checkpoint_smart(p)
    assert "embed_tokens.weight" in payload["model_state_dict"]
    assert payload["global_step"] == 7


def test_infer_config_from_state_dict(tiny_moe_config):
    torch.manual_seed(0)
    cfg0 = config_to_deepseek_config(tiny_moe_config)
    m = DeepSeekTransformer(cfg0)
    cfg = infer_config_from_state_dict(m.state_dict())
    assert cfg.vocab_size == cfg0.vocab_size
    assert cfg.hidden_size == cfg0.hidden_size
    assert cfg.num_layers == cfg0.num_layers
    assert cfg.use_moe and cfg.num_experts == cfg0.num_experts
    # roundtrip: rebuilt model accepts the state dict
    m2 = DeepSeekTransformer(cfg)
    missing, unexpected = m2.load_state_dict(m.state_dict(), strict=False)
    assert not unexpected


# ---------------------------------------------------------------- chat
def test_chat_interface_commands(small_model, tokenizer):
    chat = ChatInterface(model=small_model, tokenizer=tokenizer)
    assert "commands" in chat.handle_command("/help")
    assert chat.handle_command("/mode creative").endswith("creative")
    assert chat.gen_config.temperature == pytest.approx(1.1)
    assert chat.handle_command("/quit") == "__QUIT__"
    assert chat.handle_command("not a command") is None


def test_chat_respond_and_history(small_model, tokenizer):
    chat = ChatInterface(model=small_model, tokenizer=tokenizer)
    chat.gen_config.max_new_tokens = 4
    reply = chat.respond("hi")
    assert isinstance(reply, str)
    assert len(chat.history) == 2
    assert chat.history[0]["role"] == "user"


def test_chat_save_session(small_model, tokenizer, tmp_path):
    import json
    chat = ChatInterface(model=small_model, tokenizer=tokenizer)
    chat.gen_config.max_new_tokens = 2
    chat.respond("hello")
    p = str(tmp_path / "session.json")
    assert chat.handle_command(f"/save {p}").startswith("saved")
    data = json.loads(open(p).read())
    assert len(data["history"]) == 2


def test_merge_ep_checkpoints(tiny_moe_config, tmp_path):
    """Two EP shard files with disjoint experts merge to the full model."""
    from luminaai_amd.inference.loader import find_ep_shards, merge_ep_checkpoints
    torch.manual_seed(0)
    full = DeepSeekTransformer(config_to_deepseek_config(tiny_moe_config))
    sd = full.state_dict()
    E = tiny_moe_config.num_experts
    for r in range(2):
        shard = dict(sd)
        for k, v in sd.items():
            if ".w_gate_up" in k or ".w_down" in k:
                EL = E // 2
                shard[k] = v[r * EL:(r + 1) * EL].clone()
        torch.save({"model_state_dict": shard, "global_step": 5},
                   tmp_path / f"ck_step_5_ep_rank_{r}.pt")
    shards = find_ep_shards(str(tmp_path / "ck_step_5_ep_rank_0.pt"))
    assert len(shards) == 2
    merged = merge_ep_checkpoints(shards)
    for k, v in sd.items():
        torch.testing.assert_close(merged["model_state_dict"][k], v)
    m2 = DeepSeekTransformer(config_to_deepseek_config(tiny_moe_config))
    m2.load_state_dict(merged["model_state_dict"])


def test_wandb_fallback_logger(tmp_path, monkeypatch, tiny_config):
    monkeypatch.chdir(tmp_path)
    from luminaai_amd.monitoring import WandbLogger
    tiny_config.enable_wandb = True
    wb = WandbLogger(tiny_config, enabled=True)
    wb.log({"loss": 1.5}, step=1)
    wb.finish()
    import glob as g
    files = g.glob("experiments/*/wandb_fallback.jsonl")
    assert files, "fallback jsonl not written"


def test_static_kv_cache_matches_dynamic(small_model):
    m = small_model.eval()
    ids = torch.randint(0, 500, (1, 10))
    with torch.no_grad():
        dyn = m.make_kv_caches()
        a, _, _ = m(ids[:, :6], kv_caches=dyn)
        sta = m.make_kv_caches(max_len=16)
        b, _, _ = m(ids[:, :6], kv_caches=sta)
        torch.testing.assert_close(a, b)
        for t in range(6, 10):
            a, _, _ = m(ids[:, t:t + 1], kv_caches=dyn)
            b, _, _ = m(ids[:, t:t + 1], kv_caches=sta)
            torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-5)
        assert sta[0].k.shape[1] == 16  # preallocated, not grown


def test_prompt_truncated_to_max_context(small_model, tokenizer):
    """A prompt longer than max_context is left-truncated before prefill,
    and generation halts once the KV cache is full."""
    eng = GenerationEngine(small_model.eval(), tokenizer)
    prompt = list(range(2, 42))  # 40 tokens, max_context 16
    cfg = GenerationConfig(max_new_tokens=8, temperature=0.0, max_context=16)
    out = eng.generate(prompt, cfg)
    assert eng.get_stats()["prefill_tokens"] == 16
    # cache is already full after prefill: at most one sampled token fits
    assert len(out) <= 1


def test_stop_token_ends_generation(small_model, tokenizer):
    eng = GenerationEngine(small_model.eval(), tokenizer)
    cfg = GenerationConfig(max_new_tokens=8, temperature=0.0)
    first = eng.generate(tokenizer.encode("abc"), cfg)
    assert first
    cfg2 = GenerationConfig(max_new_tokens=8, temperature=0.0,
                            stop_token_ids=[first[0]])
    out = eng.generate(tokenizer.encode("abc"), cfg2)
    assert out == []  # greedy path re-derives first[0], which now stops


def test_load_zero_shards_merges_optimizer(tmp_path, small_model):
    """Per-rank ZeRO shards merge into one payload: model from rank0,
    flat-optimizer partitions concatenated per group."""
    from luminaai_amd.inference.loader import load_zero_shards
    sd = small_model.state_dict()
    for r in range(2):
        opt = {"groups": [{
            "exp_avg": torch.full((4,), float(r)),
            "exp_avg_sq": torch.full((4,), float(10 + r)),
            "master": torch.full((4,), float(20 + r)),
            "lr": 1e-3,
        }], "step": 5}
        torch.save({"model_state_dict": sd, "optimizer_state_dict": opt,
                    "global_step": 5},
                   tmp_path / f"optim_shard_ck_rank{r}.pt")
    merged = load_zero_shards(str(tmp_path))
    g = merged["optimizer_state_dict"]["groups"][0]
    assert g["exp_avg"].shape == (8,)
    assert float(g["exp_avg"][0]) == 0.0 and float(g["exp_avg"][4]) == 1.0
    assert float(g["master"][4]) == 21.0
    assert "embed_tokens.weight" in merged["model_state_dict"]


def test_find_latest_checkpoint(tmp_path, monkeypatch):
    from luminaai_amd.inference.loader import find_latest_checkpoint
    monkeypatch.chdir(tmp_path)
    assert find_latest_checkpoint() is None
    import os as _os
    ckdir = tmp_path / "checkpoints" / "exp"
    ckdir.mkdir(parents=True)
    torch.save({"a": 1}, ckdir / "old.pt")
    _os.utime(ckdir / "old.pt", (1000, 1000))
    torch.save({"a": 2}, ckdir / "new.pt")
    (ckdir / "skip.pt.tmp").write_text("x")
    latest = find_latest_checkpoint()
    assert latest.endswith("new.pt")


# ---------------------------------------------------------------- quantization
def test_int8_quantize_roundtrip():
    from luminaai_amd.ops.quant import dequantize_int8, quantize_int8
    torch.manual_seed(0)
    w = torch.randn(64, 128)
    q, s = quantize_int8(w)
    assert q.dtype == torch.int8 and s.shape == (64,)
    wd = dequantize_int8(q, s, torch.float32)
    # per-channel int8: worst-case error is scale/2 = amax/254 per channel
    err = (wd - w).abs().amax(dim=1)
    bound = w.abs().amax(dim=1) / 254 + 1e-6
    assert bool((err <= bound * 1.01).all())


def test_int4_quantize_roundtrip():
    from luminaai_amd.ops.quant import dequantize_int4, quantize_int4
    torch.manual_seed(0)
    w = torch.randn(32, 256)
    q, s = quantize_int4(w, group_size=128)
    assert q.dtype == torch.uint8 and q.shape == (32, 128)
    assert s.shape == (32, 2)
    wd = dequantize_int4(q, s, group_size=128, dtype=torch.float32)
    err = (wd - w).abs().reshape(32, 2, 128).amax(dim=2)
    bound = w.abs().reshape(32, 2, 128).amax(dim=2) / 14 + 1e-6
    assert bool((err <= bound * 1.01).all())


def test_quantize_model_int8_generates(tiny_moe_config, tokenizer):
    """int8-quantized model still decodes, outputs stay close to bf16."""
    from luminaai_amd.ops.quant import (quantize_model,
                                        quantized_model_bytes)
    torch.manual_seed(0)
    m = DeepSeekTransformer(config_to_deepseek_config(tiny_moe_config)).eval()
    before = quantized_model_bytes(m)
    ids = torch.randint(0, 500, (1, 12))
    with torch.no_grad():
        ref, _, _ = m(ids)
    n = quantize_model(m, mode="int8", min_dim=32)
    assert n > 0
    after = quantized_model_bytes(m)
    # embeddings and batched MoE expert weights stay full precision (the
    # fused dequant expert GEMM is a round-2 kernel); the attention/router
    # Linears shrink 4x
    assert after < before
    with torch.no_grad():
        out, _, _ = m(ids)
    # logits agree closely enough to keep the argmax most of the time
    agree = (out.argmax(-1) == ref.argmax(-1)).float().mean()
    assert float(agree) > 0.5
    eng = GenerationEngine(m, tokenizer)
    toks = eng.generate(tokenizer.encode("hi"),
                        GenerationConfig(max_new_tokens=4, temperature=0.0))
    assert all(isinstance(t, int) for t in toks)


def test_quantize_model_int4(small_model):
    from luminaai_amd.ops.quant import quantize_model
    m = small_model.eval()
    n = quantize_model(m, mode="int4", min_dim=32, group_size=32)
    assert n > 0
    ids = torch.randint(0, 500, (1, 8))
    with torch.no_grad():
        out, _, _ = m(ids)
    assert torch.isfinite(out).all()


def test_quantize_model_bad_mode(small_model):
    from luminaai_amd.ops.quant import quantize_model
    with pytest.raises(ValueError):
        quantize_model(small_model, mode="int2")


def test_generate_batch_matches_sequential(small_model, tokenizer):
    """Left-padded batched decode must reproduce per-sequence greedy decode
    exactly (dense model: no capacity interactions)."""
    eng = GenerationEngine(small_model.eval(), tokenizer)
    cfg = GenerationConfig(max_new_tokens=6, temperature=0.0,
                           stop_token_ids=[-1])
    prompts = [tokenizer.encode("hello world"),
               tokenizer.encode("a"),
               tokenizer.encode("the quick brown fox jumps")]
    seq = [eng.generate(p, cfg) for p in prompts]
    batched = eng.generate_batch(prompts, cfg)
    assert batched == seq


def test_generate_batch_per_row_configs(small_model, tokenizer):
    eng = GenerationEngine(small_model.eval(), tokenizer)
    cfgs = [GenerationConfig(max_new_tokens=2, temperature=0.0,
                             stop_token_ids=[-1]),
            GenerationConfig(max_new_tokens=5, temperature=0.0,
                             stop_token_ids=[-1])]
    outs = eng.generate_batch([tokenizer.encode("abc"),
                               tokenizer.encode("defgh")], cfgs)
    assert len(outs[0]) <= 2 and len(outs[1]) <= 5
    # row 0's shorter budget must not truncate row 1
    solo = eng.generate(tokenizer.encode("defgh"), cfgs[1])
    assert outs[1] == solo


def test_generate_batch_moe_model(tiny_moe_config, tokenizer):
    torch.manual_seed(0)
    m = DeepSeekTransformer(config_to_deepseek_config(tiny_moe_config)).eval()
    eng = GenerationEngine(m, tokenizer)
    outs = eng.generate_batch([tokenizer.encode("hi"),
                               tokenizer.encode("something longer")],
                              GenerationConfig(max_new_tokens=4,
                                               temperature=0.0))
    assert len(outs) == 2
    assert all(isinstance(t, int) for o in outs for t in o)


@pytest.mark.parametrize("draft_k", [1, 3, 5])
def test_speculative_matches_greedy(small_model, tokenizer, draft_k):
    """Speculative output must equal plain target greedy decode for ANY
    draft — here a different random model of the same vocab."""
    from luminaai_amd.models.transformer import DeepSeekConfig
    torch.manual_seed(9)
    draft = DeepSeekTransformer(DeepSeekConfig(
        vocab_size=512, hidden_size=32, num_layers=1, num_heads=2,
        num_kv_heads=1, intermediate_size=64, seq_length=64, use_moe=False,
        use_mod=False, tie_word_embeddings=False)).eval()
    eng = GenerationEngine(small_model.eval(), tokenizer)
    cfg = GenerationConfig(max_new_tokens=8, temperature=0.0,
                           stop_token_ids=[-1])
    for prompt in ("hello there", "x"):
        p = tokenizer.encode(prompt)
        ref = eng.generate(p, cfg)
        spec = eng.generate_speculative(p, draft, cfg, draft_k=draft_k)
        assert spec == ref, (draft_k, prompt)


def test_speculative_self_draft_accepts_everything(small_model, tokenizer):
    """Target drafting for itself accepts every proposal (sanity on the
    acceptance bookkeeping)."""
    eng = GenerationEngine(small_model.eval(), tokenizer)
    cfg = GenerationConfig(max_new_tokens=6, temperature=0.0,
                           stop_token_ids=[-1])
    p = tokenizer.encode("self draft")
    assert eng.generate_speculative(p, small_model, cfg, draft_k=3) == \
        eng.generate(p, cfg)


def test_speculative_sampling_runs_on_real_model(small_model, tokenizer):
    """Sampling-mode speculative decode on a real transformer pair."""
    from luminaai_amd.models.transformer import DeepSeekConfig
    torch.manual_seed(4)
    draft = DeepSeekTransformer(DeepSeekConfig(
        vocab_size=512, hidden_size=32, num_layers=1, num_heads=2,
        num_kv_heads=1, intermediate_size=64, seq_length=64, use_moe=False,
        use_mod=False, tie_word_embeddings=False)).eval()
    eng = GenerationEngine(small_model.eval(), tokenizer)
    cfg = GenerationConfig(max_new_tokens=6, temperature=0.9, top_k=50,
                           top_p=0.95, stop_token_ids=[-1])
    out = eng.generate_speculative(tokenizer.encode("abc"), draft, cfg,
                                   draft_k=3)
    assert 0 < len(out) <= 6
    assert all(0 <= t < 512 for t in out)


def test_speculative_sampling_preserves_distribution(tokenizer):
    """Acceptance-rejection speculative sampling must reproduce the
    target's sampling distribution for any draft. Fixed-logit stub models
    make the target distribution analytic; compare empirically."""
    import torch.nn as nn

    V = 8

    class _StubCache:
        def __init__(self):
            self._len = 0

        @property
        def seq_len(self):
            return self._len

        def truncate(self, n):
            self._len = min(self._len, n)

    class Stub(nn.Module):
        """Position-independent logits; KV cache only tracks length."""

        def __init__(self, logits):
            super().__init__()
            self.logits = logits
            self.dummy = nn.Parameter(torch.zeros(1))

        def make_kv_caches(self, max_len=0):
            self._caches = [_StubCache()]
            return self._caches

        def forward(self, ids, kv_caches=None, **kw):
            B, S = ids.shape
            if kv_caches is not None:
                kv_caches[0]._len += S
            out = self.logits.view(1, 1, V).expand(B, S, V)
            return out, None, None

    torch.manual_seed(0)
    t_logits = torch.tensor([2.0, 1.5, 1.0, 0.5, 0.0, -0.5, -1.0, -8.0])
    d_logits = torch.tensor([0.0, 0.0, 2.0, 2.0, 0.0, 1.0, -1.0, -8.0])
    target, draft = Stub(t_logits), Stub(d_logits)

    from luminaai_amd.inference.engine import (GenerationConfig,
                                               GenerationEngine)
    cfg = GenerationConfig(max_new_tokens=3, temperature=1.0, top_k=0,
                           top_p=1.0, repetition_penalty=1.0,
                           stop_token_ids=[-1])
    eng = GenerationEngine(target, tokenizer, torch.device("cpu"))
    # expected = plain softmax of target logits (no filters active);
    # token V-1 has ~0 mass and the tokenizer eos (id 256+) is out of range
    expected = torch.softmax(t_logits, -1)
    counts = torch.zeros(V)
    n_runs = 400
    torch.manual_seed(1234)
    for _ in range(n_runs):
        toks = eng.generate_speculative([1, 2], draft, cfg, draft_k=2)
        for t in toks:
            counts[t] += 1
    emp = counts / counts.sum()
    tv = 0.5 * float((emp - expected).abs().sum())
    assert tv < 0.06, (tv, emp.tolist(), expected.tolist())


def test_quantize_moe_experts_int8(tiny_moe_config, tokenizer):
    """Expert weights (the bulk of a MoE model) drop to int8 storage and
    the model still decodes with close logits."""
    from luminaai_amd.ops.quant import (quantize_model,
                                        quantized_model_bytes)
    torch.manual_seed(0)
    m = DeepSeekTransformer(config_to_deepseek_config(tiny_moe_config)).eval()
    before = quantized_model_bytes(m)
    ids = torch.randint(0, 500, (1, 12))
    with torch.no_grad():
        ref, _, _ = m(ids)
    n = quantize_model(m, mode="int8", min_dim=32)
    assert n > 0
    assert not hasattr(m.layers[1].ffn, "w_gate_up")  # Parameter removed
    assert m.layers[1].ffn.w_gate_up_q.dtype == torch.int8
    after = quantized_model_bytes(m)
    assert after < before * 0.62   # experts fp32->int8 dominate the drop
    with torch.no_grad():
        out, _, _ = m(ids)
    agree = (out.argmax(-1) == ref.argmax(-1)).float().mean()
    assert float(agree) > 0.5
    eng = GenerationEngine(m, tokenizer)
    toks = eng.generate(tokenizer.encode("hello"),
                        GenerationConfig(max_new_tokens=4, temperature=0.0))
    assert all(isinstance(t, int) for t in toks)


def test_chat_with_draft_model(small_model, tokenizer):
    """Chat replies through speculative decoding when a draft is set, and
    greedy replies match the non-speculative path exactly."""
    from luminaai_amd.models.transformer import DeepSeekConfig
    chat = ChatInterface(model=small_model, tokenizer=tokenizer)
    chat.gen_config = GenerationConfig(max_new_tokens=5, temperature=0.0,
                                       stop_token_ids=[-1])
    plain = chat.respond("hello there")
    chat.history.clear()
    torch.manual_seed(2)
    draft = DeepSeekTransformer(DeepSeekConfig(
        vocab_size=512, hidden_size=32, num_layers=1, num_heads=2,
        num_kv_heads=1, intermediate_size=64, seq_length=64, use_moe=False,
        use_mod=False, tie_word_embeddings=False)).eval()
    chat.set_draft_model(draft)
    spec = chat.respond("hello there")
    assert spec == plain


def test_speculative_with_quantized_target(small_model, tokenizer):
    """int8 target + fp draft still decodes (quantization changes the
    target's argmax sequence, so just check validity + determinism)."""
    import copy
    from luminaai_amd.models.transformer import DeepSeekConfig
    from luminaai_amd.ops.quant import quantize_model
    target = copy.deepcopy(small_model).eval()
    quantize_model(target, mode="int8", min_dim=32)
    torch.manual_seed(5)
    draft = DeepSeekTransformer(DeepSeekConfig(
        vocab_size=512, hidden_size=32, num_layers=1, num_heads=2,
        num_kv_heads=1, intermediate_size=64, seq_length=64, use_moe=False,
        use_mod=False, tie_word_embeddings=False)).eval()
    eng = GenerationEngine(target, tokenizer)
    cfg = GenerationConfig(max_new_tokens=5, temperature=0.0,
                           stop_token_ids=[-1])
    a = eng.generate_speculative(tokenizer.encode("q"), draft, cfg)
    b = eng.generate(tokenizer.encode("q"), cfg)
    assert a == b  # speculative == plain greedy on the SAME (int8) target


def test_chat_run_loop(small_model, tokenizer, monkeypatch, capsys):
    """The REPL loop: commands, a chat turn, and /quit."""
    chat = ChatInterface(model=small_model, tokenizer=tokenizer)
    chat.gen_config.max_new_tokens = 2
    lines = iter(["/help", "", "hello there", "/stats", "/quit"])
    monkeypatch.setattr("builtins.input", lambda *_: next(lines))
    chat.run()
    out = capsys.readouterr().out
    assert "commands:" in out
    assert "ai> " in out
    assert "tokens_generated" in out
    assert len(chat.history) == 2


def test_quantized_model_state_dict_roundtrip(tiny_moe_config, tokenizer):
    """int8 buffers serialize: a quantized model's state dict reloads into
    a freshly-quantized skeleton and produces identical logits."""
    from luminaai_amd.ops.quant import quantize_model
    torch.manual_seed(0)
    m = DeepSeekTransformer(config_to_deepseek_config(tiny_moe_config)).eval()
    quantize_model(m, mode="int8", min_dim=32)
    ids = torch.randint(0, 500, (1, 8))
    with torch.no_grad():
        ref, _, _ = m(ids)
    sd = m.state_dict()
    torch.manual_seed(1)   # different init to prove the load matters
    m2 = DeepSeekTransformer(config_to_deepseek_config(tiny_moe_config)).eval()
    quantize_model(m2, mode="int8", min_dim=32)
    m2.load_state_dict(sd)
    with torch.no_grad():
        out, _, _ = m2(ids)
    torch.testing.assert_close(out, ref)


def test_int4_group_size_constraint():
    import torch.nn as nn
    from luminaai_amd.ops.quant import Int4Linear, quantize_model
    with pytest.raises(AssertionError):
        Int4Linear(100, 32, group_size=64)   # 100 % 64 != 0
    # quantize_model silently skips incompatible layers instead
    m = nn.Sequential(nn.Linear(100, 64, bias=False))
    assert quantize_model(m, mode="int4", min_dim=32, group_size=64) == 0
    assert type(m[0]) is nn.Linear


def test_generation_mode_fallback():
    g = GenerationConfig.from_mode("no_such_mode")
    assert g.temperature == pytest.approx(0.8)   # standard


def test_server_gen_config_parsing():
    from luminaai_amd.inference.server import _gen_config
    cfg = _gen_config({"mode": "precise", "max_tokens": 7,
                       "stop_token_ids": [5, 9],
                       "repetition_penalty": 1.3})
    assert cfg.max_new_tokens == 7
    assert cfg.temperature == pytest.approx(0.3)
    assert cfg.stop_token_ids == [5, 9]
    assert cfg.repetition_penalty == pytest.approx(1.3)
