"""Checkpoint manager, format-compatible with the reference
(/root/reference/Src/Main_Scripts/training/checkpoint.py:14-362): torch.save
dict with keys model_state_dict / optimizer_state_dict / scheduler_state_dict /
global_step / epoch / config / model_config / timestamps, plus best-symlink,
history JSON, retention and emergency save."""

from __future__ import annotations

import dataclasses
import json
import os
import shutil
import time
from pathlib import Path
from typing import Dict, List, Optional

import torch


class CheckpointManager:
    def __init__(self, checkpoint_dir: str, save_total_limit: int = 5):
        self.dir = Path(checkpoint_dir)
        self.dir.mkdir(parents=True, exist_ok=True)
        self.save_total_limit = save_total_limit
        self.history_path = self.dir / "checkpoint_history.json"
        self.history: List[Dict] = []
        if self.history_path.exists():
            try:
                self.history = json.loads(self.history_path.read_text())
            except json.JSONDecodeError:
                self.history = []

    # ------------------------------------------------------------------
    def save_checkpoint(self, model, optimizer=None, scheduler=None,
                        global_step: int = 0, epoch: int = 0,
                        config=None, model_config=None,
                        metrics: Optional[Dict] = None,
                        is_best: bool = False, tag: Optional[str] = None) -> str:
        name = tag or f"checkpoint_step_{global_step}"
        path = self.dir / f"{name}.pt"
        payload = {
            "model_state_dict": _unwrap(model).state_dict(),
            "optimizer_state_dict": optimizer.state_dict() if optimizer else None,
            "scheduler_state_dict": scheduler.state_dict() if scheduler else None,
            "global_step": global_step,
            "epoch": epoch,
            "config": _to_dict(config),
            "model_config": _to_dict(model_config),
            "metrics": metrics or {},
            "timestamp": time.time(),
            "framework": "luminaai_amd",
        }
        tmp = path.with_suffix(".tmp")
        torch.save(payload, tmp)
        os.replace(tmp, path)

        self.history.append({
            "path": str(path), "global_step": global_step, "epoch": epoch,
            "metrics": metrics or {}, "timestamp": payload["timestamp"],
        })
        self._write_history()
        if is_best:
            best = self.dir / "best_checkpoint.pt"
            if best.is_symlink() or best.exists():
                best.unlink()
            try:
                best.symlink_to(path.name)
            except OSError:
                shutil.copy2(path, best)
        self._cleanup_old_checkpoints()
        return str(path)

    # ------------------------------------------------------------------
    def load_checkpoint(self, which: str = "latest", map_location="cpu") -> Dict:
        path = self.resolve(which)
        if path is None:
            raise FileNotFoundError(f"no checkpoint matching '{which}' in {self.dir}")
        # corrupted-checkpoint fallback: latest -> previous -> best
        candidates = [path] + [Path(h["path"]) for h in reversed(self.history)
                               if Path(h["path"]) != path]
        last_err = None
        for p in candidates:
            if not p.exists():
                continue
            try:
                return torch.load(p, map_location=map_location, weights_only=False)
            except Exception as e:  # noqa: BLE001 - any unreadable ckpt falls through
                last_err = e
                continue
        raise RuntimeError(f"all checkpoints unreadable: {last_err}")

    def resolve(self, which: str) -> Optional[Path]:
        if which == "best":
            p = self.dir / "best_checkpoint.pt"
            return p if p.exists() else None
        if which == "latest":
            if self.history:
                return Path(self.history[-1]["path"])
            pts = sorted(self.dir.glob("checkpoint_*.pt"),
                         key=lambda p: p.stat().st_mtime)
            return pts[-1] if pts else None
        p = Path(which)
        return p if p.exists() else (self.dir / which if (self.dir / which).exists() else None)

    # ------------------------------------------------------------------
    def validate_compatibility(self, payload: Dict, model) -> bool:
        """Shape-compare the stored state dict against the model
        (reference checkpoint.py:315-353)."""
        sd = payload.get("model_state_dict", {})
        msd = _unwrap(model).state_dict()
        for k, v in sd.items():
            if k in msd and msd[k].shape != v.shape:
                return False
        return True

    def emergency_save(self, model, global_step: int = 0, **kw) -> str:
        return self.save_checkpoint(model, global_step=global_step,
                                    tag=f"emergency_step_{global_step}", **kw)

    def _cleanup_old_checkpoints(self):
        keep_paths = set()
        best = self.dir / "best_checkpoint.pt"
        if best.is_symlink():
            keep_paths.add(str((self.dir / os.readlink(best)).resolve()))
        regular = [h for h in self.history if "emergency" not in h["path"]]
        excess = len(regular) - self.save_total_limit
        if excess <= 0:
            return
        removed = []
        for h in regular:
            if excess <= 0:
                break
            p = Path(h["path"])
            if str(p.resolve()) in keep_paths:
                continue
            if p.exists():
                p.unlink()
            removed.append(h)
            excess -= 1
        self.history = [h for h in self.history if h not in removed]
        self._write_history()

    def reload_history(self):
        """Re-read the on-disk history (ranks that never wrote the main
        checkpoint share the directory but have an empty in-memory list)."""
        if self.history_path.exists():
            try:
                self.history = json.loads(self.history_path.read_text())
            except json.JSONDecodeError:
                pass
        return self.history

    def _write_history(self):
        self.history_path.write_text(json.dumps(self.history, indent=2))


def _unwrap(model):
    return model.module if hasattr(model, "module") else model


def _to_dict(obj):
    if obj is None:
        return None
    if dataclasses.is_dataclass(obj):
        return dataclasses.asdict(obj)
    if hasattr(obj, "to_dict"):
        return obj.to_dict()
    if isinstance(obj, dict):
        return obj
    return {k: v for k, v in vars(obj).items() if not k.startswith("_")}
