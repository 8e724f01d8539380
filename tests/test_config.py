"""Config preset integrity and roundtrip tests."""

import pytest

from luminaai_amd.config import Config, ConfigPresets, ConfigManager


def test_all_presets_construct_and_validate():
    for name in ConfigPresets.names():
        cfg = ConfigPresets.get(name)
        cfg.validate()
        assert cfg.hidden_size % cfg.num_heads == 0, name
        assert cfg.num_heads % cfg.num_kv_heads == 0, name
        assert cfg.estimate_total_params() > 0, name
        assert cfg.estimate_active_params() <= cfg.estimate_total_params(), name


def test_headline_preset_geometry():
    """The b1 MoE headline config matches the reference preset
    (config_manager.py:1007-1065: 1908 hidden / 31 layers / 12H / 4KV,
    seq 2048, 8 experts top-2)."""
    c = ConfigPresets.b1_moe()
    assert (c.hidden_size, c.num_layers, c.num_heads, c.num_kv_heads) == \
        (1908, 31, 12, 4)
    assert c.seq_length == 2048
    assert c.use_moe and c.num_experts == 8 and c.moe_top_k == 2
    assert not c.use_mod
    total = c.estimate_total_params()
    active = c.estimate_active_params()
    assert 6e9 < total < 12e9, f"~8B total expected, got {total/1e9:.1f}B"
    assert 1e9 < active < 3e9, f"~1.3B active expected, got {active/1e9:.1f}B"


def test_yaml_roundtrip(tmp_path):
    c = ConfigPresets.debug()
    c.learning_rate = 3.25e-4
    p = str(tmp_path / "c.yaml")
    c.save(p)
    c2 = Config.load(p)
    assert c2.learning_rate == pytest.approx(3.25e-4)
    assert c2.hidden_size == c.hidden_size
    assert ConfigManager.validate_config(c2)


def test_auto_configure():
    c = Config(hidden_size=512, num_layers=4, num_heads=8, num_kv_heads=4,
               vocab_size=1000, intermediate_size=None)
    assert c.vocab_size % 64 == 0          # padded up
    assert c.intermediate_size is not None
    assert c.intermediate_size % 256 == 0


def test_param_and_memory_estimators():
    """Estimators track the real built model within ~15%."""
    from luminaai_amd.config import ConfigPresets
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    cfg = ConfigPresets.debug()
    cfg.use_moe = True
    cfg.num_experts = 4
    cfg.moe_top_k = 2
    est_total = cfg.estimate_total_params()
    est_active = cfg.estimate_active_params()
    m = DeepSeekTransformer(config_to_deepseek_config(cfg))
    real_total = m.count_parameters()
    real_active = m.count_active_parameters()
    assert abs(est_total - real_total) / real_total < 0.15, \
        (est_total, real_total)
    assert abs(est_active - real_active) / real_active < 0.20, \
        (est_active, real_active)
    assert est_active <= est_total
    assert cfg.estimate_memory_gb() > 0


def test_largest_divisor_helper():
    from luminaai_amd.config import largest_divisor_leq
    assert largest_divisor_leq(8, 8) == 8
    assert largest_divisor_leq(8, 5) == 4
    assert largest_divisor_leq(12, 5) == 4
    assert largest_divisor_leq(7, 3) == 1


def test_estimator_respects_moe_pattern():
    """hybrid_70b (MoE every 2nd layer): the estimator must count only
    the MoE layers' expert stacks (round-2 fix: it assumed pattern=all
    and over-counted hybrid_70b ~2x)."""
    import torch
    from luminaai_amd.config import ConfigPresets
    from luminaai_amd.models import (DeepSeekTransformer,
                                     config_to_deepseek_config)
    cfg = ConfigPresets.hybrid_70b()
    with torch.device("meta"):
        m = DeepSeekTransformer(config_to_deepseek_config(cfg))
    real = sum(p.numel() for p in m.parameters())
    est = cfg.estimate_total_params()
    assert abs(est - real) / real < 0.01, (est, real)
