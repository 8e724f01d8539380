"""Chinchilla-scaling auto-epoch logic.

Rebuild of the reference chinchilla scaler
(/root/reference/Src/Main_Scripts/training/chinchilla_scaler.py:38-568:
ConvergenceDetector :38, ComputeEfficiencyTracker :109,
AdaptiveCurriculumManager :155, EnhancedChinchillaScaler :177).
"""

from __future__ import annotations

import json
import math
import time
from collections import deque
from typing import Dict, List, Optional


class ConvergenceDetector:
    """Plateau/divergence/convergence scoring over a rolling loss window
    (reference chinchilla_scaler.py:38-107)."""

    def __init__(self, window: int = 50, plateau_std: float = 0.01,
                 divergence_delta: float = 0.3):
        self.window = window
        self.plateau_std = plateau_std
        self.divergence_delta = divergence_delta
        self.losses: deque = deque(maxlen=window)

    def update(self, loss: float):
        if loss == loss and not math.isinf(loss):
            self.losses.append(loss)

    def _stats(self):
        n = len(self.losses)
        if n < 2:
            return None
        mean = sum(self.losses) / n
        var = sum((x - mean) ** 2 for x in self.losses) / n
        return mean, math.sqrt(var)

    def is_plateaued(self) -> bool:
        if len(self.losses) < self.window:
            return False
        s = self._stats()
        return s is not None and s[1] < self.plateau_std

    def is_diverging(self) -> bool:
        if len(self.losses) < 10:
            return False
        recent = list(self.losses)[-5:]
        earlier = list(self.losses)[:5]
        return (sum(recent) / len(recent)
                - sum(earlier) / len(earlier)) > self.divergence_delta

    def convergence_score(self) -> float:
        """0 (diverging) .. 1 (fully converged/plateaued)."""
        if len(self.losses) < 4:
            return 0.0
        half = len(self.losses) // 2
        first = sum(list(self.losses)[:half]) / half
        second = sum(list(self.losses)[half:]) / (len(self.losses) - half)
        if first <= 0:
            return 0.0
        improvement = (first - second) / abs(first)
        return max(0.0, min(1.0, 1.0 - improvement * 10.0))


class ComputeEfficiencyTracker:
    """FLOPs accounting: 6·params per token; loss reduction per FLOP
    (reference chinchilla_scaler.py:109-153)."""

    def __init__(self, n_params: int):
        self.n_params = n_params
        self.tokens_processed = 0
        self.initial_loss: Optional[float] = None
        self.latest_loss: Optional[float] = None
        self.t_start = time.time()

    def update(self, tokens: int, loss: float):
        self.tokens_processed += tokens
        if self.initial_loss is None:
            self.initial_loss = loss
        self.latest_loss = loss

    @property
    def total_flops(self) -> float:
        return 6.0 * self.n_params * self.tokens_processed

    def loss_reduction_per_exaflop(self) -> float:
        if self.initial_loss is None or self.latest_loss is None or \
                self.total_flops == 0:
            return 0.0
        return (self.initial_loss - self.latest_loss) / (self.total_flops / 1e18)

    def report(self) -> Dict:
        return {
            "tokens_processed": self.tokens_processed,
            "total_flops": self.total_flops,
            "flops_per_sec": self.total_flops / max(time.time() - self.t_start, 1e-9),
            "loss_reduction_per_exaflop": self.loss_reduction_per_exaflop(),
        }


class AdaptiveCurriculumManager:
    """Sequence-length curriculum: ramp from min_frac of seq_length to full
    over warmup_frac of training (reference chinchilla_scaler.py:155-176)."""

    def __init__(self, seq_length: int, min_frac: float = 0.25,
                 warmup_frac: float = 0.1):
        self.seq_length = seq_length
        self.min_frac = min_frac
        self.warmup_frac = warmup_frac

    def seq_len_at(self, progress: float) -> int:
        """progress in [0,1] -> current curriculum sequence length (multiple of 64)."""
        if progress >= self.warmup_frac:
            return self.seq_length
        frac = self.min_frac + (1 - self.min_frac) * (progress / self.warmup_frac)
        return max(64, int(self.seq_length * frac) // 64 * 64)


class EnhancedChinchillaScaler:
    """optimal_tokens = multiplier × params; epochs = ceil(optimal / dataset
    tokens), clamped [1, 50] (reference chinchilla_scaler.py:177-560)."""

    def __init__(self, config, model=None, dataset_tokens: Optional[int] = None,
                 multiplier: float = 20.0):
        self.config = config
        self.multiplier = multiplier
        if model is not None:
            self.n_params = sum(p.numel() for p in model.parameters())
        else:
            self.n_params = config.estimate_total_params()
        self.dataset_tokens = dataset_tokens or 0
        self.convergence = ConvergenceDetector()
        self.efficiency = ComputeEfficiencyTracker(self.n_params)
        self.curriculum = AdaptiveCurriculumManager(config.seq_length)
        self.optimal_tokens = self.multiplier * self.n_params
        self.recommended_epochs = 1

    def compute_optimal_epochs(self, dataset_tokens: Optional[int] = None) -> int:
        if dataset_tokens:
            self.dataset_tokens = dataset_tokens
        if self.dataset_tokens <= 0:
            return 1
        self.recommended_epochs = max(
            1, min(50, math.ceil(self.optimal_tokens / self.dataset_tokens)))
        return self.recommended_epochs

    def update(self, tokens: int, loss: float):
        self.convergence.update(loss)
        self.efficiency.update(tokens, loss)

    def should_stop_early(self) -> bool:
        """Stop when converged AND the Chinchilla-optimal token budget is
        reached, or on divergence past the budget (reference :377-404)."""
        budget_done = self.efficiency.tokens_processed >= self.optimal_tokens
        if budget_done and self.convergence.is_plateaued():
            return True
        return budget_done and self.convergence.is_diverging()

    def progress(self) -> float:
        if self.optimal_tokens <= 0:
            return 1.0
        return min(1.0, self.efficiency.tokens_processed / self.optimal_tokens)

    def status_report(self) -> Dict:
        return {
            "n_params": self.n_params,
            "optimal_tokens": self.optimal_tokens,
            "dataset_tokens": self.dataset_tokens,
            "recommended_epochs": self.recommended_epochs,
            "progress": self.progress(),
            "convergence_score": self.convergence.convergence_score(),
            "plateaued": self.convergence.is_plateaued(),
            "diverging": self.convergence.is_diverging(),
            "efficiency": self.efficiency.report(),
        }

    def save_state(self, path: str):
        with open(path, "w") as f:
            json.dump({
                "tokens_processed": self.efficiency.tokens_processed,
                "losses": list(self.convergence.losses),
                "recommended_epochs": self.recommended_epochs,
            }, f)

    def load_state(self, path: str):
        with open(path) as f:
            st = json.load(f)
        self.efficiency.tokens_processed = st.get("tokens_processed", 0)
        for x in st.get("losses", []):
            self.convergence.update(x)
        self.recommended_epochs = st.get("recommended_epochs", 1)
