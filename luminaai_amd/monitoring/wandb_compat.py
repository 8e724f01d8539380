"""Optional Weights & Biases logging (reference trainer.py:44-50,
enable_wandb config flag). wandb is not installed in the offline image, so
this wrapper degrades to JSON-lines logging with the same call surface."""

from __future__ import annotations

import json
import os
import time
from typing import Dict, Optional

try:
    import wandb as _wandb
    _HAS_WANDB = True
except ImportError:
    _wandb = None
    _HAS_WANDB = False


class WandbLogger:
    def __init__(self, config, enabled: Optional[bool] = None):
        self.enabled = (enabled if enabled is not None
                        else getattr(config, "enable_wandb", False))
        self.run = None
        self._fallback_path = None
        if not self.enabled:
            return
        if _HAS_WANDB:
            self.run = _wandb.init(
                project=getattr(config, "wandb_project", None) or "luminaai-amd",
                entity=getattr(config, "wandb_entity", None),
                name=getattr(config, "experiment_name", None),
                config=config.to_dict() if hasattr(config, "to_dict") else None)
        else:
            exp = getattr(config, "experiment_name", None) or "default"
            d = os.path.join("experiments", exp)
            os.makedirs(d, exist_ok=True)
            self._fallback_path = os.path.join(d, "wandb_fallback.jsonl")

    def log(self, metrics: Dict, step: Optional[int] = None):
        if not self.enabled:
            return
        if self.run is not None:
            self.run.log(metrics, step=step)
        elif self._fallback_path:
            rec = {"step": step, "ts": time.time(),
                   **{k: v for k, v in metrics.items()
                      if isinstance(v, (int, float, str))}}
            with open(self._fallback_path, "a") as f:
                f.write(json.dumps(rec) + "\n")

    def finish(self):
        if self.run is not None:
            self.run.finish()
