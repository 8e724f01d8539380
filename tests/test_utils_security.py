"""Tests for utils (environment/reporting/data/profiling) and security."""

import os
import time

import pytest

from luminaai_amd.security import (InputValidator, RateLimiter,
                                   SecureConversationalChat, SecurityManager)
from luminaai_amd.utils import (create_training_report, estimate_training_time,
                                flatten_conversation_tree,
                                generate_sample_data, get_profiling_stats,
                                get_system_info, profile_function,
                                profiling_context, reset_profiling_stats,
                                validate_environment, validate_jsonl)
from luminaai_amd.utils.profiling import enable_profiling


# ---------------------------------------------------------------- environment
def test_system_info_and_validate(tiny_config):
    info = get_system_info()
    assert info["torch"] and info["cpu_count"] > 0
    v = validate_environment(tiny_config)
    assert isinstance(v["ok"], bool)
    assert "info" in v


def test_estimate_training_time(tiny_config):
    est = estimate_training_time(tiny_config, dataset_tokens=10 ** 9, n_gpus=8)
    assert est["est_seconds"] > 0
    est1 = estimate_training_time(tiny_config, dataset_tokens=10 ** 9, n_gpus=1)
    assert est1["est_seconds"] == pytest.approx(est["est_seconds"] * 8)


# ---------------------------------------------------------------- profiling
def test_profiling_decorator_and_context():
    reset_profiling_stats()
    enable_profiling(True)

    @profile_function("unit_test_fn")
    def f():
        time.sleep(0.01)
        return 42

    assert f() == 42
    with profiling_context("unit_test_ctx"):
        time.sleep(0.005)
    stats = get_profiling_stats()
    enable_profiling(False)
    assert stats["unit_test_fn"]["calls"] == 1
    assert stats["unit_test_fn"]["total_s"] >= 0.005
    assert "unit_test_ctx" in stats


def test_profiling_disabled_is_noop():
    reset_profiling_stats()
    enable_profiling(False)

    @profile_function("should_not_appear")
    def f():
        return 1

    f()
    assert "should_not_appear" not in get_profiling_stats()


# ---------------------------------------------------------------- data utils
def test_validate_jsonl_and_sample_gen(tmp_path):
    p = str(tmp_path / "sample.jsonl")
    generate_sample_data(p, n=10)
    stats = validate_jsonl(p)
    assert stats["valid"] == 10 and stats["ok"]
    assert stats["roles"]["user"] == 10

    bad = tmp_path / "bad.jsonl"
    bad.write_text('{"messages": [{"role": "user", "content": "x"}]}\n'
                   "not json\n"
                   '{"nothing": true}\n')
    stats = validate_jsonl(str(bad))
    assert stats["valid"] == 1 and len(stats["errors"]) == 2


def test_flatten_conversation_tree():
    tree = {"prompt": {"role": "user", "text": "q", "replies": [
        {"role": "assistant", "text": "a1", "replies": []},
        {"role": "assistant", "text": "a2", "replies": [
            {"role": "user", "text": "follow", "replies": []}]},
    ]}}
    convs = flatten_conversation_tree(tree)
    assert len(convs) == 2
    assert convs[0]["messages"][0]["content"] == "q"
    assert len(convs[1]["messages"]) == 3


def test_training_report(tmp_path, tiny_config):
    history = [{"epoch": 0, "mean_loss": 2.5, "tokens_per_sec": 1000.0,
                "duration_s": 10.0, "eval": {"loss": 2.4}}]
    p = create_training_report(history, tiny_config,
                               str(tmp_path / "report.html"))
    text = open(p).read()
    assert "2.5000" in text and "Training report" in text


# ---------------------------------------------------------------- security
def test_auth_roundtrip():
    sm = SecurityManager(secret_key="k" * 32)
    assert sm.register_user("alice", "correcthorse")
    assert not sm.register_user("alice", "again1234")  # duplicate
    assert not sm.register_user("bob", "short")        # too short
    tok = sm.authenticate("alice", "correcthorse")
    assert tok and sm.validate_session(tok) == "alice"
    assert sm.authenticate("alice", "wrongpass") is None
    assert sm.validate_session("garbage.token") is None


def test_auth_lockout():
    sm = SecurityManager()
    sm.register_user("carol", "password123")
    for _ in range(5):
        sm.authenticate("carol", "nope-nope")
    assert sm.is_locked_out("carol")
    assert sm.authenticate("carol", "password123") is None  # locked out


def test_session_expiry():
    sm = SecurityManager(session_ttl=-1)
    sm.register_user("dave", "password123")
    tok = sm.authenticate("dave", "password123")
    assert sm.validate_session(tok) is None


def test_rate_limiter():
    rl = RateLimiter({"message": (3, 60.0)})
    assert all(rl.allow("u", "message") for _ in range(3))
    assert not rl.allow("u", "message")
    assert rl.remaining("u", "message") == 0
    rl.reset("u")
    assert rl.allow("u", "message")


def test_input_validator():
    v = InputValidator(max_chars=100)
    assert v.validate("hello there")["ok"]
    assert not v.validate("<script>alert(1)</script>")["ok"]
    assert not v.validate("a" * 200)["ok"]
    assert not v.validate("")["ok"]
    assert not v.validate("1; DROP TABLE users")["ok"]
    assert "<" not in v.sanitize("<b>hi</b>")


def test_secure_chat_wrapper(small_model, tokenizer):
    from luminaai_amd.inference import ChatInterface
    chat = ChatInterface(model=small_model, tokenizer=tokenizer)
    chat.gen_config.max_new_tokens = 2
    sec = SecureConversationalChat(chat, user="eve",
                                   rate_limiter=RateLimiter({"message": (2, 60.0)}))
    assert isinstance(sec.respond("hi"), str)
    assert sec.respond("<script>x</script>").startswith("[input rejected")
    sec.respond("ok")
    assert sec.respond("third").startswith("[rate limited")


def test_chrome_trace_export(tmp_path):
    import json
    from luminaai_amd.utils.profiling import (enable_profiling,
                                              export_chrome_trace,
                                              profile_function,
                                              profiling_context,
                                              reset_profiling_stats)
    reset_profiling_stats()
    enable_profiling(True, trace=True)
    try:
        @profile_function("traced_fn")
        def f():
            return sum(range(1000))

        for _ in range(3):
            f()
        with profiling_context("traced_ctx"):
            f()
        p = str(tmp_path / "trace.json")
        n = export_chrome_trace(p)
        assert n >= 5
        data = json.loads(open(p).read())
        names = {e["name"] for e in data["traceEvents"]}
        assert {"traced_fn", "traced_ctx"} <= names
        for e in data["traceEvents"]:
            assert e["ph"] == "X" and e["dur"] >= 0 and e["ts"] >= 0
    finally:
        enable_profiling(False)
        reset_profiling_stats()


def test_module_profilers(small_model):
    import torch
    from luminaai_amd.utils.profiling import (attach_module_profilers,
                                              enable_profiling,
                                              get_profiling_stats,
                                              reset_profiling_stats)
    reset_profiling_stats()
    enable_profiling(True)
    handles = attach_module_profilers(small_model)
    try:
        with torch.no_grad():
            small_model(torch.randint(0, 500, (1, 8)))
        stats = get_profiling_stats()
        layer_keys = [k for k in stats if k.startswith("layer")]
        assert len(layer_keys) == len(small_model.layers)
        assert all(stats[k]["calls"] == 1 for k in layer_keys)
    finally:
        for h in handles:
            h.remove()
        enable_profiling(False)
        reset_profiling_stats()
