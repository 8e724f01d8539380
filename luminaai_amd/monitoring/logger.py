"""Metrics collection, health monitoring and structured logging.

Rebuild of the reference monitoring layer
(/root/reference/Src/Main_Scripts/monitoring/logger.py:29-640:
MetricsCollector :29, TrainingHealthMonitor :276). `ProductionLogger` is
provided here for real — the reference orchestrator imported it
(orchestrator.py:683) but it was never defined.
"""

from __future__ import annotations

import json
import logging
import math
import os
import time
from collections import deque
from typing import Dict, List, Optional


DEFAULT_THRESHOLDS = {
    "loss_max": 20.0,
    "grad_norm_max": 100.0,
    "lr_min": 1e-8,
    "lr_max": 1.0,
    "memory_frac_max": 0.95,
}


class MetricsCollector:
    """Rolling metric windows with threshold alerting and a 0-100 health
    score (reference logger.py:29-274)."""

    def __init__(self, window: int = 200,
                 thresholds: Optional[Dict[str, float]] = None):
        self.window = window
        self.thresholds = dict(DEFAULT_THRESHOLDS, **(thresholds or {}))
        self.series: Dict[str, deque] = {}
        self.alerts: List[Dict] = []

    def log(self, name: str, value: float, step: Optional[int] = None):
        if value is None or value != value:
            return
        self.series.setdefault(name, deque(maxlen=self.window)).append(
            (step, float(value)))
        self._check_threshold(name, float(value), step)

    def log_many(self, metrics: Dict[str, float], step: Optional[int] = None):
        for k, v in metrics.items():
            if isinstance(v, (int, float)):
                self.log(k, v, step)

    def _check_threshold(self, name: str, value: float, step):
        alert = None
        if name == "loss" and value > self.thresholds["loss_max"]:
            alert = f"loss {value:.2f} above {self.thresholds['loss_max']}"
        elif name == "grad_norm" and value > self.thresholds["grad_norm_max"]:
            alert = f"grad_norm {value:.1f} above {self.thresholds['grad_norm_max']}"
        elif name == "lr" and not (self.thresholds["lr_min"] <= value
                                   <= self.thresholds["lr_max"]):
            alert = f"lr {value:.2e} out of range"
        elif name == "memory_frac" and value > self.thresholds["memory_frac_max"]:
            alert = f"memory {value:.0%} above {self.thresholds['memory_frac_max']:.0%}"
        if alert:
            self.alerts.append({"metric": name, "message": alert,
                                "step": step, "timestamp": time.time()})

    def latest(self, name: str) -> Optional[float]:
        s = self.series.get(name)
        return s[-1][1] if s else None

    def mean(self, name: str, last_n: Optional[int] = None) -> Optional[float]:
        s = self.series.get(name)
        if not s:
            return None
        vals = [v for _, v in s][-last_n:] if last_n else [v for _, v in s]
        return sum(vals) / len(vals)

    def trend(self, name: str, last_n: int = 50) -> Optional[float]:
        """Least-squares slope over the last n points
        (reference logger.py:223-245)."""
        s = self.series.get(name)
        if not s or len(s) < 3:
            return None
        ys = [v for _, v in s][-last_n:]
        n = len(ys)
        xs = range(n)
        mx, my = (n - 1) / 2.0, sum(ys) / n
        num = sum((x - mx) * (y - my) for x, y in zip(xs, ys))
        den = sum((x - mx) ** 2 for x in xs)
        return num / den if den else 0.0

    def health_score(self) -> float:
        """0-100 composite (reference logger.py:246-274)."""
        score = 100.0
        loss_trend = self.trend("loss")
        if loss_trend is not None and loss_trend > 0:
            score -= min(30.0, loss_trend * 1000.0)
        gn = self.latest("grad_norm")
        if gn is not None:
            if gn > 100:
                score -= 30
            elif gn > 10:
                score -= 10
        recent_alerts = [a for a in self.alerts
                         if time.time() - a["timestamp"] < 300]
        score -= min(30.0, 5.0 * len(recent_alerts))
        return max(0.0, min(100.0, score))


class TrainingHealthMonitor:
    """Phase tracking + periodic health checks + recommendations
    (reference logger.py:276-633)."""

    PHASES = ("warmup", "early", "middle", "late")

    def __init__(self, total_steps: Optional[int] = None,
                 warmup_steps: int = 0, check_every: int = 100):
        self.collector = MetricsCollector()
        self.total_steps = total_steps
        self.warmup_steps = warmup_steps
        self.check_every = check_every
        self.step = 0
        self.reports: List[Dict] = []

    def log_step(self, metrics: Dict[str, float], step: Optional[int] = None):
        self.step = step if step is not None else self.step + 1
        self.collector.log_many(metrics, self.step)
        if self.check_every and self.step % self.check_every == 0:
            self.reports.append(self.health_check())

    def phase(self) -> str:
        if self.step < self.warmup_steps:
            return "warmup"
        if not self.total_steps:
            return "early"
        frac = self.step / self.total_steps
        if frac < 0.25:
            return "early"
        if frac < 0.75:
            return "middle"
        return "late"

    def health_check(self) -> Dict:
        score = self.collector.health_score()
        recs = []
        gn = self.collector.latest("grad_norm")
        loss_trend = self.collector.trend("loss")
        if gn is not None and gn > 50:
            recs.append("consider lowering the learning rate (high grad norm)")
        if loss_trend is not None and loss_trend > 0 and self.phase() != "warmup":
            recs.append("loss trending up — check data or reduce LR")
        if loss_trend is not None and abs(loss_trend) < 1e-5 and \
                self.phase() == "late":
            recs.append("loss plateaued in late phase — consider stopping")
        return {"step": self.step, "phase": self.phase(), "health_score": score,
                "recommendations": recs,
                "alerts": self.collector.alerts[-5:],
                "timestamp": time.time()}

    def save_report(self, path: str):
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        with open(path, "w") as f:
            json.dump({"final": self.health_check(),
                       "history": self.reports[-50:]}, f, indent=2)


class ProductionLogger:
    """Structured stdlib logger writing console + timestamped JSON-lines
    file (the reference referenced this class but never shipped it,
    orchestrator.py:683)."""

    def __init__(self, name: str = "luminaai", log_dir: Optional[str] = None,
                 level: int = logging.INFO):
        self.logger = logging.getLogger(name)
        self.logger.setLevel(level)
        if not self.logger.handlers:
            ch = logging.StreamHandler()
            ch.setFormatter(logging.Formatter(
                "%(asctime)s %(levelname)s %(name)s: %(message)s"))
            self.logger.addHandler(ch)
        self.jsonl_path = None
        if log_dir:
            os.makedirs(log_dir, exist_ok=True)
            self.jsonl_path = os.path.join(
                log_dir, f"train_{int(time.time())}.jsonl")

    def __getattr__(self, item):
        return getattr(self.logger, item)

    def log_metrics(self, metrics: Dict, step: Optional[int] = None):
        rec = {"step": step, "timestamp": time.time(), **{
            k: v for k, v in metrics.items() if isinstance(v, (int, float, str))}}
        if self.jsonl_path:
            with open(self.jsonl_path, "a") as f:
                f.write(json.dumps(rec) + "\n")
