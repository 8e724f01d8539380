import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a ROCm GPU (run on MI355X)")
    config.addinivalue_line("markers", "slow: long-running test")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tiny_config():
    from luminaai_amd.config import Config
    return Config(
        vocab_size=512, hidden_size=64, num_layers=2, num_heads=4,
        num_kv_heads=2, seq_length=64, intermediate_size=128,
        batch_size=2, micro_batch_size=2, gradient_accumulation_steps=1,
        num_epochs=1, num_workers=0, use_moe=False, use_mod=False,
        zero_stage=0, eval_every_n_batches=0, save_every_n_batches=0,
        gradient_checkpointing=False, experiment_name="test_run",
        precision="fp32",
    )


@pytest.fixture
def tiny_moe_config(tiny_config):
    c = tiny_config
    c.use_moe = True
    c.num_experts = 4
    c.moe_top_k = 2
    return c


@pytest.fixture
def tokenizer():
    from luminaai_amd.data.tokenizer import ConversationTokenizer
    return ConversationTokenizer(max_length=64)


@pytest.fixture
def small_model(tiny_config):
    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    torch.manual_seed(0)
    return DeepSeekTransformer(config_to_deepseek_config(tiny_config))


@pytest.fixture
def sample_conversations(tmp_path):
    import json
    p = tmp_path / "conv.jsonl"
    rows = [
        {"messages": [{"role": "user", "content": f"question {i}?"},
                      {"role": "assistant", "content": f"answer {i}."}]}
        for i in range(8)
    ]
    p.write_text("\n".join(json.dumps(r) for r in rows))
    return str(p)


@pytest.fixture
def sample_text(tmp_path):
    p = tmp_path / "base.txt"
    p.write_text("the quick brown fox jumps over the lazy dog. " * 200)
    return str(p)
