"""Adaptive training orchestrator (control plane).

Rebuild of the reference orchestrator
(/root/reference/Src/Main_Scripts/training/orchestrator.py:79-2163:
MetaLearningEngine :79, AdaptiveHyperparameterOptimizer :303,
ArchitectureEvolution :389, RealTimeAnalytics :453, ProductionMonitoring :630,
AdaptiveTrainingOrchestrator :673).

Differences from the reference, on purpose:
- the trainer exposes a metrics hook (`Trainer.set_metrics_hook`), so the
  orchestrator subscribes instead of monkey-patching `train_step` /
  `optimizer_step` (reference orchestrator.py:1265-1456);
- `initialize_training` wires the CALLER's model/trainer when given — the
  reference silently built a second model and trained that one instead of
  the backend-wrapped model from Main (orchestrator.py:1153-1170);
- ProductionMonitoring drift/toxicity scores are honest None-stubs rather
  than `np.random.random()` placeholders (orchestrator.py:638-672).
"""

from __future__ import annotations

import json
import math
import os
import queue
import threading
import time
from collections import deque
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .trainer import Trainer, TrainingMetrics


# ======================================================================
@dataclass
class AdaptiveDecision:
    """One intervention decided by the control plane
    (reference orchestrator.py:69-78)."""
    action: str                  # lr_adjust | emergency_lr | plateau_lr |
                                 # divergence_lr | rollback | add_expert |
                                 # prune_expert | batch_size | mod_capacity
    value: Optional[float] = None
    reason: str = ""
    confidence: float = 1.0
    timestamp: float = field(default_factory=time.time)


# ======================================================================
class MetaLearningEngine:
    """Records run outcomes and suggests hyperparameters from similar past
    runs (reference orchestrator.py:79-302). State is JSON, not pickle."""

    def __init__(self, state_path: str = "experiments/meta_learning.json"):
        self.state_path = state_path
        self.runs: List[Dict] = []
        self._load()

    def _load(self):
        if os.path.exists(self.state_path):
            try:
                with open(self.state_path) as f:
                    self.runs = json.load(f).get("runs", [])
            except (json.JSONDecodeError, OSError):
                self.runs = []

    def save(self):
        os.makedirs(os.path.dirname(self.state_path) or ".", exist_ok=True)
        with open(self.state_path, "w") as f:
            json.dump({"runs": self.runs[-200:]}, f)

    def record_run(self, config, final_loss: float, steps: int,
                   interventions: int):
        self.runs.append({
            "hidden_size": config.hidden_size,
            "num_layers": config.num_layers,
            "use_moe": config.use_moe,
            "num_experts": config.num_experts if config.use_moe else 0,
            "learning_rate": config.learning_rate,
            "batch": (config.micro_batch_size or 1)
            * config.gradient_accumulation_steps,
            "final_loss": final_loss,
            "steps": steps,
            "interventions": interventions,
            "timestamp": time.time(),
        })
        self.save()

    def _similarity(self, run: Dict, config) -> float:
        s = 0.0
        s += 1.0 / (1.0 + abs(math.log(max(run["hidden_size"], 1)
                                       / max(config.hidden_size, 1))))
        s += 1.0 if run["use_moe"] == config.use_moe else 0.0
        s += 1.0 / (1.0 + abs(run["num_layers"] - config.num_layers) / 8.0)
        return s

    def suggest_hyperparameters(self, config) -> Optional[Dict]:
        """Best-loss hyperparams among the most-similar past runs."""
        if not self.runs:
            return None
        scored = sorted(self.runs,
                        key=lambda r: self._similarity(r, config), reverse=True)
        top = [r for r in scored[:5] if r["final_loss"] == r["final_loss"]]
        if not top:
            return None
        best = min(top, key=lambda r: r["final_loss"])
        return {"learning_rate": best["learning_rate"], "batch": best["batch"],
                "expected_loss": best["final_loss"]}


# ======================================================================
class AdaptiveHyperparameterOptimizer:
    """LR / batch-size policy with per-decision cooldown
    (reference orchestrator.py:303-388: >=50-step cooldown; plateau
    std<0.01 -> LRx1.5; divergence delta>0.3 -> x0.5; steady progress ->
    x1.2; grad-norm mean>10 -> x0.7)."""

    def __init__(self, cooldown_steps: int = 50):
        self.cooldown_steps = cooldown_steps
        self.last_decision_step = -10 ** 9
        self.losses: deque = deque(maxlen=100)
        self.grad_norms: deque = deque(maxlen=100)

    def observe(self, m: TrainingMetrics):
        if m.loss is not None and m.loss == m.loss:
            self.losses.append(m.loss)
        if m.grad_norm is not None and m.grad_norm == m.grad_norm:
            self.grad_norms.append(m.grad_norm)

    def propose(self, step: int, current_lr: float) -> Optional[AdaptiveDecision]:
        if step - self.last_decision_step < self.cooldown_steps:
            return None
        if len(self.losses) < 20:
            return None
        window = list(self.losses)[-20:]
        mean = sum(window) / len(window)
        std = math.sqrt(sum((x - mean) ** 2 for x in window) / len(window))
        delta = window[-1] - window[0]
        decision = None
        if self.grad_norms and \
                sum(self.grad_norms) / len(self.grad_norms) > 10.0:
            decision = AdaptiveDecision("lr_adjust", current_lr * 0.7,
                                        "high mean grad norm", 0.8)
        elif delta > 0.3:
            decision = AdaptiveDecision("divergence_lr", current_lr * 0.5,
                                        f"loss rising by {delta:.3f}", 0.9)
        elif std < 0.01:
            decision = AdaptiveDecision("plateau_lr", current_lr * 1.5,
                                        f"plateau std={std:.4f}", 0.6)
        elif delta < -0.05:
            decision = AdaptiveDecision("lr_adjust", current_lr * 1.2,
                                        "steady progress", 0.5)
        if decision is not None:
            self.last_decision_step = step
        return decision

    def propose_batch_size(self, memory_gb: float, total_gb: float,
                           micro_batch: int) -> Optional[AdaptiveDecision]:
        """Memory-driven batch-size change (reference orchestrator.py:367-388)."""
        if total_gb <= 0:
            return None
        util = memory_gb / total_gb
        if util > 0.92 and micro_batch > 1:
            return AdaptiveDecision("batch_size", micro_batch // 2,
                                    f"memory {util:.0%}", 0.9)
        if util < 0.5 and micro_batch < 512:
            return AdaptiveDecision("batch_size", micro_batch * 2,
                                    f"memory only {util:.0%}", 0.4)
        return None


# ======================================================================
class ArchitectureEvolution:
    """Expert add/prune policy (reference orchestrator.py:389-452)."""

    def __init__(self, min_experts: int = 2, max_experts: int = 128,
                 cooldown_steps: int = 500):
        self.min_experts = min_experts
        self.max_experts = max_experts
        self.cooldown_steps = cooldown_steps
        self.last_change_step = -10 ** 9

    def propose(self, step: int, expert_stats: Optional[Dict]) \
            -> Optional[AdaptiveDecision]:
        if expert_stats is None or \
                step - self.last_change_step < self.cooldown_steps:
            return None
        util = expert_stats.get("mean_utilization", 1.0)
        imb = expert_stats.get("max_imbalance", 1.0)
        n = expert_stats.get("num_experts", 0)
        decision = None
        if util > 0.95 and imb > 4.0 and n < self.max_experts:
            decision = AdaptiveDecision(
                "add_expert", None,
                f"all experts saturated (util={util:.2f}, imb={imb:.1f})", 0.6)
        elif util < 0.5 and n > self.min_experts:
            decision = AdaptiveDecision(
                "prune_expert", None,
                f"under-utilised experts (util={util:.2f})", 0.7)
        elif imb > 2.0:
            decision = AdaptiveDecision(
                "load_balance", None,
                f"EP load imbalance {imb:.1f}", 0.7)
        if decision is not None:
            self.last_change_step = step
        return decision


# ======================================================================
class RealTimeAnalytics:
    """Anomaly detection + convergence prediction
    (reference orchestrator.py:453-628: loss spike = mean+2*std and
    delta>0.1; grad explosion = >100 or >10x historical mean; expert
    collapse; quadratic loss-curve fit)."""

    def __init__(self, window: int = 100):
        self.losses: deque = deque(maxlen=window)
        self.grad_norms: deque = deque(maxlen=window)

    def observe(self, m: TrainingMetrics):
        if m.loss is not None and m.loss == m.loss:
            self.losses.append(m.loss)
        if m.grad_norm is not None and m.grad_norm == m.grad_norm:
            self.grad_norms.append(m.grad_norm)

    def detect_anomalies(self, m: TrainingMetrics) -> List[Dict]:
        anomalies = []
        if m.loss is not None and (m.loss != m.loss or math.isinf(m.loss)):
            anomalies.append({"type": "nan_loss", "severity": "critical"})
        if len(self.losses) >= 10 and m.loss is not None:
            hist = list(self.losses)[:-1]
            mean = sum(hist) / len(hist)
            std = math.sqrt(sum((x - mean) ** 2 for x in hist) / len(hist))
            if m.loss > mean + 2 * std and m.loss - mean > 0.1:
                anomalies.append({"type": "loss_spike", "severity": "high",
                                  "value": m.loss, "mean": mean})
        if m.grad_norm is not None and len(self.grad_norms) >= 5:
            hist_mean = sum(self.grad_norms) / len(self.grad_norms)
            if m.grad_norm > 100.0 or \
                    (hist_mean > 0 and m.grad_norm > 10.0 * hist_mean):
                anomalies.append({"type": "grad_explosion", "severity": "high",
                                  "value": m.grad_norm})
        es = m.expert_stats
        if es and es.get("mean_entropy", 1.0) < 0.1:
            anomalies.append({"type": "expert_collapse", "severity": "medium",
                              "entropy": es["mean_entropy"]})
        return anomalies

    def predict_convergence(self) -> Optional[Dict]:
        """Least-squares quadratic fit of the loss window; predicted floor
        and steps-to-floor (reference orchestrator.py:479-553)."""
        n = len(self.losses)
        if n < 20:
            return None
        ys = list(self.losses)
        xs = list(range(n))
        # normal equations for y = a x^2 + b x + c
        s0, s1, s2, s3, s4 = n, sum(xs), sum(x * x for x in xs), \
            sum(x ** 3 for x in xs), sum(x ** 4 for x in xs)
        t0 = sum(ys)
        t1 = sum(x * y for x, y in zip(xs, ys))
        t2 = sum(x * x * y for x, y in zip(xs, ys))
        det = (s4 * (s2 * s0 - s1 * s1) - s3 * (s3 * s0 - s1 * s2)
               + s2 * (s3 * s1 - s2 * s2))
        if abs(det) < 1e-12:
            return None
        a = (t2 * (s2 * s0 - s1 * s1) - s3 * (t1 * s0 - s1 * t0)
             + s2 * (t1 * s1 - s2 * t0)) / det
        b = (s4 * (t1 * s0 - t0 * s1) - t2 * (s3 * s0 - s1 * s2)
             + s2 * (s3 * t0 - t1 * s2)) / det
        c = (s4 * (s2 * t0 - t1 * s1) - s3 * (s3 * t0 - t1 * s2)
             + t2 * (s3 * s1 - s2 * s2)) / det
        out = {"trend": "improving" if ys[-1] < ys[0] else "worsening",
               "fit": {"a": a, "b": b, "c": c}}
        if a > 1e-12 and b < 0:
            x_min = -b / (2 * a)
            out["predicted_floor"] = a * x_min ** 2 + b * x_min + c
            out["steps_to_floor"] = max(0.0, x_min - (n - 1))
        return out


# ======================================================================
class ProductionMonitoring:
    """Serving-side quality tracking slots (reference orchestrator.py:630-672
    had np.random placeholders for drift/toxicity; here the scores are
    honest None until a real scorer is registered)."""

    def __init__(self):
        self.drift_scorer = None
        self.toxicity_scorer = None
        self.history: deque = deque(maxlen=1000)

    def register_scorers(self, drift=None, toxicity=None):
        self.drift_scorer = drift
        self.toxicity_scorer = toxicity

    def score_sample(self, text: str) -> Dict:
        rec = {
            "semantic_drift": self.drift_scorer(text) if self.drift_scorer else None,
            "toxicity": self.toxicity_scorer(text) if self.toxicity_scorer else None,
            "timestamp": time.time(),
        }
        self.history.append(rec)
        return rec


# ======================================================================
class AdaptiveTrainingOrchestrator:
    """Owns trainer + background monitor thread + decision loop
    (reference orchestrator.py:673-2163)."""

    def __init__(self, config, model=None, tokenizer=None,
                 trainer: Optional[Trainer] = None, logger=None):
        self.config = config
        self.logger = logger
        self._ext_model = model
        self._ext_tokenizer = tokenizer
        self.trainer = trainer
        self.meta = MetaLearningEngine(
            os.path.join("experiments", config.experiment_name or "default",
                         "meta_learning.json"))
        self.hyperopt = AdaptiveHyperparameterOptimizer()
        self.evolution = ArchitectureEvolution()
        self.analytics = RealTimeAnalytics()
        self.production = ProductionMonitoring()
        self.metrics_queue: "queue.Queue[TrainingMetrics]" = queue.Queue(maxsize=1000)
        self._monitor_thread: Optional[threading.Thread] = None
        self._stop_event = threading.Event()
        self.decisions: List[AdaptiveDecision] = []
        self.interventions_executed = 0
        self._train_result: Optional[Dict] = None

    # ---------------------------------------------------------------- setup
    def initialize_training(self):
        """Build (or adopt) tokenizer/model/trainer and subscribe to its
        metric stream."""
        if self.trainer is None:
            from ..data.tokenizer import ConversationTokenizer
            from ..models import DeepSeekTransformer, config_to_deepseek_config
            tok = self._ext_tokenizer or ConversationTokenizer(
                max_length=self.config.seq_length)
            model = self._ext_model or DeepSeekTransformer(
                config_to_deepseek_config(self.config))
            self.trainer = Trainer(model, tok, self.config, logger=self.logger)
        self.trainer.set_metrics_hook(self._enqueue_metrics)
        if self.config.enable_adaptive_lr:
            sug = self.meta.suggest_hyperparameters(self.config)
            if sug and self.logger:
                self.logger.info(f"meta-learning suggestion: {sug}")
        return self.trainer

    def _enqueue_metrics(self, m: TrainingMetrics):
        try:
            self.metrics_queue.put_nowait(m)
        except queue.Full:      # drop-oldest (reference orchestrator.py:1290-1297)
            try:
                self.metrics_queue.get_nowait()
                self.metrics_queue.put_nowait(m)
            except (queue.Empty, queue.Full):
                pass

    # ---------------------------------------------------------------- monitor
    def start_real_time_monitoring(self):
        if self._monitor_thread is not None:
            return
        self._stop_event.clear()
        self._monitor_thread = threading.Thread(
            target=self._monitor_loop, daemon=True, name="lumina-monitor")
        self._monitor_thread.start()

    def stop_monitoring(self):
        self._stop_event.set()
        if self._monitor_thread is not None:
            self._monitor_thread.join(timeout=5.0)
            self._monitor_thread = None

    def _monitor_loop(self):
        # Fixed 50 ms drain cadence instead of per-item queue wakeups: at
        # millisecond step times a wakeup per emitted metric costs a GIL
        # handoff per step (measured 36% at debug scale in round 2's first
        # pass); decision latency stays far below the adaptive cooldowns
        # (>= 50 steps, reference orchestrator.py:312).
        while not self._stop_event.is_set():
            self._stop_event.wait(0.05)
            while True:
                try:
                    m = self.metrics_queue.get_nowait()
                except queue.Empty:
                    break
                try:
                    self._process_metrics(m)
                except Exception:  # noqa: BLE001 — must not kill training
                    if self.logger:
                        self.logger.exception("monitor error")

    def _process_metrics(self, m: TrainingMetrics):
        self.analytics.observe(m)
        self.hyperopt.observe(m)
        for anomaly in self.analytics.detect_anomalies(m):
            self._handle_anomaly(anomaly, m)
        if self.trainer is None:
            return
        d = self.hyperopt.propose(m.step or 0, m.lr or 0.0)
        if d:
            self._execute_decision(d)
        d = self.evolution.propose(m.step or 0, m.expert_stats)
        if d:
            self._execute_decision(d)

    def _handle_anomaly(self, anomaly: Dict, m: TrainingMetrics):
        kind = anomaly["type"]
        if kind == "nan_loss":
            self._execute_decision(AdaptiveDecision(
                "rollback", 100, "NaN/Inf loss", 1.0))
        elif kind == "loss_spike":
            self._execute_decision(AdaptiveDecision(
                "emergency_lr", 0.1, f"loss spike to {anomaly['value']:.3f}", 0.9))
        elif kind == "grad_explosion":
            self._execute_decision(AdaptiveDecision(
                "emergency_lr", 0.5, f"grad norm {anomaly['value']:.1f}", 0.8))
        elif kind == "expert_collapse":
            self._execute_decision(AdaptiveDecision(
                "routing_temperature", 1.5, "expert collapse", 0.7))

    # ---------------------------------------------------------------- dispatch
    def _execute_decision(self, d: AdaptiveDecision) -> bool:
        """Map decision -> trainer intervention
        (reference orchestrator.py:1040-1131)."""
        t = self.trainer
        if t is None:
            return False
        ok = False
        if d.action in ("lr_adjust", "plateau_lr", "divergence_lr"):
            ok = t.adjust_learning_rate(float(d.value))
        elif d.action == "emergency_lr":
            t.emergency_lr_reduction(float(d.value))
            ok = True
        elif d.action == "rollback":
            ok = t.rollback_steps(int(d.value or 100))
        elif d.action == "add_expert":
            ok = t.add_expert()
        elif d.action == "prune_expert":
            ok = t.prune_expert()
        elif d.action == "load_balance":
            ok = t.apply_expert_load_balance()
        elif d.action == "batch_size":
            t.adjust_batch_size(int(d.value))
            ok = True
        elif d.action == "mod_capacity":
            t.adjust_mod_capacity(float(d.value))
            ok = True
        elif d.action == "routing_temperature":
            t.adjust_routing_temperature(float(d.value))
            ok = True
        self.decisions.append(d)
        if ok:
            self.interventions_executed += 1
            if self.logger:
                self.logger.info(f"adaptive: {d.action}={d.value} ({d.reason})")
        return ok

    # ---------------------------------------------------------------- run
    def run_adaptive_training(self, train_dataset=None, eval_dataset=None,
                              train_loader=None, eval_loader=None) -> Dict:
        if self.trainer is None:
            self.initialize_training()
        self.start_real_time_monitoring()
        try:
            result = self.trainer.train(train_dataset, eval_dataset,
                                        train_loader, eval_loader)
        finally:
            self.stop_monitoring()
        self._train_result = result
        final = self.trainer._metric_floats().get("ce_loss", float("nan"))
        self.meta.record_run(self.config, final, self.trainer.global_step,
                             self.interventions_executed)
        return result

    # ---------------------------------------------------------------- status
    def get_adaptive_status(self) -> Dict:
        return {
            "monitoring": self._monitor_thread is not None
            and self._monitor_thread.is_alive(),
            "queue_depth": self.metrics_queue.qsize(),
            "decisions": len(self.decisions),
            "interventions_executed": self.interventions_executed,
            "recent_decisions": [
                {"action": d.action, "value": d.value, "reason": d.reason}
                for d in self.decisions[-5:]],
            "convergence": self.analytics.predict_convergence(),
        }

    def cleanup(self):
        self.stop_monitoring()
        self.meta.save()


# Fallback simple driver (reference AdaptiveTrainer, orchestrator.py:1971)
class AdaptiveTrainer:
    def __init__(self, config, **kw):
        self.orchestrator = AdaptiveTrainingOrchestrator(config, **kw)

    def train(self, *a, **kw):
        self.orchestrator.initialize_training()
        return self.orchestrator.run_adaptive_training(*a, **kw)
