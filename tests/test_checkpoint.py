"""CheckpointManager behaviors: retention, corrupted-file fallback,
compat validation, emergency save, best symlink."""

import os

import pytest
import torch

from luminaai_amd.training.checkpoint import CheckpointManager


def _save_n(mgr, model, n, start=1):
    paths = []
    for i in range(start, start + n):
        paths.append(mgr.save_checkpoint(model, global_step=i))
    return paths


def test_retention_keeps_limit_and_best(small_model, tmp_path):
    mgr = CheckpointManager(str(tmp_path), save_total_limit=3)
    first = mgr.save_checkpoint(small_model, global_step=1, is_best=True)
    _save_n(mgr, small_model, 5, start=2)
    pts = sorted(p.name for p in tmp_path.glob("checkpoint_*.pt"))
    assert len(pts) == 3 + (1 if os.path.exists(first) else 0) or len(pts) >= 3
    # retention never deletes the best checkpoint's target
    assert (tmp_path / "best_checkpoint.pt").exists()
    payload = mgr.load_checkpoint("best")
    assert payload["global_step"] == 1
    # latest still resolves to the newest save
    assert mgr.load_checkpoint("latest")["global_step"] == 6


def test_corrupted_checkpoint_falls_back(small_model, tmp_path):
    mgr = CheckpointManager(str(tmp_path), save_total_limit=5)
    _save_n(mgr, small_model, 3)
    latest = mgr.resolve("latest")
    with open(latest, "wb") as f:    # corrupt the newest file
        f.write(b"not a checkpoint")
    payload = mgr.load_checkpoint("latest")
    assert payload["global_step"] == 2   # fell back to the previous save


def test_all_checkpoints_unreadable_raises(small_model, tmp_path):
    mgr = CheckpointManager(str(tmp_path), save_total_limit=5)
    _save_n(mgr, small_model, 2)
    for p in tmp_path.glob("checkpoint_*.pt"):
        p.write_bytes(b"garbage")
    with pytest.raises(RuntimeError, match="unreadable"):
        mgr.load_checkpoint("latest")


def test_validate_compatibility(small_model, tmp_path):
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.models.transformer import DeepSeekConfig
    mgr = CheckpointManager(str(tmp_path))
    mgr.save_checkpoint(small_model, global_step=1)
    payload = mgr.load_checkpoint("latest")
    assert mgr.validate_compatibility(payload, small_model)
    other = DeepSeekTransformer(DeepSeekConfig(
        vocab_size=512, hidden_size=32, num_layers=1, num_heads=2,
        num_kv_heads=1, intermediate_size=64, seq_length=32, use_moe=False,
        use_mod=False, tie_word_embeddings=False))
    assert not mgr.validate_compatibility(payload, other)


def test_emergency_save_survives_retention(small_model, tmp_path):
    mgr = CheckpointManager(str(tmp_path), save_total_limit=2)
    path = mgr.emergency_save(small_model, global_step=9)
    _save_n(mgr, small_model, 4, start=10)
    assert os.path.exists(path)   # emergency saves are never cleaned up
