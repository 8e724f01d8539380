"""Continuous (iteration-level) batching tests: mid-flight joins must
reproduce solo greedy decoding exactly."""

import pytest
import torch

from luminaai_amd.inference.continuous import ContinuousBatchingEngine
from luminaai_amd.inference.engine import GenerationConfig, GenerationEngine


def _cfg(n=6):
    return GenerationConfig(max_new_tokens=n, temperature=0.0,
                            stop_token_ids=[-1])


def test_continuous_matches_solo_staggered(small_model, tokenizer):
    """Requests joining a RUNNING batch at different steps decode exactly
    what they would solo (dense model, greedy)."""
    solo = GenerationEngine(small_model.eval(), tokenizer)
    prompts = [tokenizer.encode("the first and longest prompt here"),
               tokenizer.encode("second one"),
               tokenizer.encode("third")]
    refs = [solo.generate(p, _cfg()) for p in prompts]

    eng = ContinuousBatchingEngine(small_model.eval(), tokenizer,
                                   max_batch=4, max_len=64)
    outs = eng.run_to_completion(prompts, _cfg(),
                                 admit_schedule=[0, 2, 4])
    assert outs == refs
    assert eng.stats["admitted"] == 3
    assert eng.stats["finished"] == 3


def test_continuous_more_requests_than_slots(small_model, tokenizer):
    """Slot reuse: 5 requests through a 2-slot pool, all exact."""
    solo = GenerationEngine(small_model.eval(), tokenizer)
    prompts = [tokenizer.encode(f"prompt number {i} with some words")
               for i in range(5)]
    refs = [solo.generate(p, _cfg(4)) for p in prompts]
    eng = ContinuousBatchingEngine(small_model.eval(), tokenizer,
                                   max_batch=2, max_len=64)
    outs = eng.run_to_completion(prompts, _cfg(4))
    assert outs == refs


def test_continuous_streaming_callbacks(small_model, tokenizer):
    eng = ContinuousBatchingEngine(small_model.eval(), tokenizer,
                                   max_batch=2, max_len=64)
    seen = []
    b = eng.admit(tokenizer.encode("stream me"), _cfg(5),
                  stream_callback=seen.append)
    while eng.slots[b].active or eng.slots[b].result is None:
        eng.step()
    assert seen == eng.slots[b].result
    assert len(seen) <= 5


def test_continuous_admission_gating(small_model, tokenizer):
    """A long prompt cannot join a live short batch until the cursor
    passes it; it becomes admissible as decode advances."""
    eng = ContinuousBatchingEngine(small_model.eval(), tokenizer,
                                   max_batch=2, max_len=64)
    eng.admit(tokenizer.encode("ab"), _cfg(20))
    long_prompt = tokenizer.encode("a much longer prompt than two tokens")
    assert not eng.can_admit(len(long_prompt))
    for _ in range(len(long_prompt) + 2):
        eng.step()
    assert eng.can_admit(len(long_prompt))
    b = eng.admit(long_prompt, _cfg(3))
    solo = GenerationEngine(small_model.eval(), tokenizer)
    ref = solo.generate(long_prompt, _cfg(3))
    while eng.slots[b].active or eng.slots[b].result is None:
        eng.step()
    assert eng.slots[b].result == ref


def test_continuous_empty_batch_resets_cursor(small_model, tokenizer):
    eng = ContinuousBatchingEngine(small_model.eval(), tokenizer,
                                   max_batch=2, max_len=32)
    b = eng.admit(tokenizer.encode("hello"), _cfg(3))
    while eng.slots[b].active:
        eng.step()
    assert eng.n_active() == 0
    cur_before = eng.cursor
    assert cur_before > 0
    eng.admit(tokenizer.encode("again"), _cfg(2))
    # the cache was reclaimed before the new prefill
    assert eng.cursor == len(tokenizer.encode("again"))


def test_continuous_compaction_equivalence(small_model, tokenizer):
    """Partially-occupied pool (compacted forward, batch=len(live)) must
    match a fully-occupied pool's output for the same request."""
    p = tokenizer.encode("compaction check prompt")
    solo = GenerationEngine(small_model.eval(), tokenizer)
    ref = solo.generate(p, _cfg(5))
    # pool of 8 slots, single request -> compacted path (B=1) throughout
    eng = ContinuousBatchingEngine(small_model.eval(), tokenizer,
                                   max_batch=8, max_len=64)
    out = eng.run_to_completion([p], _cfg(5))
    assert out == [ref]


def test_continuous_with_quantized_model(tiny_moe_config, tokenizer):
    """Serving combos: int8-quantized MoE model through continuous
    batching."""
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    from luminaai_amd.ops.quant import quantize_model
    torch.manual_seed(0)
    m = DeepSeekTransformer(config_to_deepseek_config(tiny_moe_config)).eval()
    quantize_model(m, mode="int8", min_dim=32)
    eng = ContinuousBatchingEngine(m, tokenizer, max_batch=2, max_len=48)
    outs = eng.run_to_completion([tokenizer.encode("quantized"),
                                  tokenizer.encode("serving")], _cfg(4))
    assert len(outs) == 2
    assert all(isinstance(t, int) for o in outs for t in o)
