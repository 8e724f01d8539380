"""Prometheus exporter for training and serving metrics.

Production-deployment observability beyond the reference (its
ProductionLogger import never resolved, reference orchestrator.py:683).
Training: `PrometheusExporter` subscribes to the trainer's metrics hook
(same queue the orchestrator consumes) and exposes loss / LR / grad-norm /
throughput / MoE balance as gauges on an HTTP endpoint. Serving:
`server_metrics()` hands the server counters/histograms.

Import-guarded: everything degrades to no-ops if prometheus_client is
missing.
"""

from __future__ import annotations

from typing import Optional

try:
    from prometheus_client import (CollectorRegistry, Counter, Gauge,
                                   Histogram, generate_latest,
                                   start_http_server)
    HAVE_PROM = True
except ImportError:  # pragma: no cover
    HAVE_PROM = False


class PrometheusExporter:
    """Feed trainer metrics to a Prometheus scrape endpoint.

    Usage:
        exporter = PrometheusExporter(port=9500)   # port=None: no server,
        trainer.set_metrics_hook(exporter)         # scrape via registry
    Chain with the orchestrator by passing `next_hook`.
    """

    def __init__(self, port: Optional[int] = None, next_hook=None,
                 registry=None):
        if not HAVE_PROM:
            raise ImportError("prometheus_client not available")
        self.registry = registry or CollectorRegistry()
        self.next_hook = next_hook
        g = lambda name, doc: Gauge(name, doc, registry=self.registry)  # noqa: E731
        self.loss = g("lumina_train_loss", "training loss")
        self.lr = g("lumina_learning_rate", "learning rate")
        self.grad_norm = g("lumina_grad_norm", "global gradient norm")
        self.tokens_per_sec = g("lumina_tokens_per_sec", "training throughput")
        self.step = g("lumina_global_step", "optimizer step")
        self.epoch = g("lumina_epoch", "epoch")
        self.expert_imbalance = g("lumina_expert_imbalance",
                                  "MoE max/mean expert load")
        self.memory_gb = g("lumina_gpu_memory_gb", "allocated GPU memory")
        self.steps_total = Counter("lumina_steps_total",
                                   "optimizer steps observed",
                                   registry=self.registry)
        if port is not None:
            start_http_server(port, registry=self.registry)

    def __call__(self, m):
        """Metrics-hook entry (TrainingMetrics from the trainer)."""
        try:
            if getattr(m, "loss", None) is not None:
                self.loss.set(float(m.loss))
            # TrainingMetrics uses `lr` / `memory_gb`; accept both spellings
            lr = getattr(m, "lr", None)
            if lr is None:
                lr = getattr(m, "learning_rate", None)
            if lr is not None:
                self.lr.set(float(lr))
            if getattr(m, "grad_norm", None) is not None:
                self.grad_norm.set(float(m.grad_norm))
            if getattr(m, "tokens_per_sec", None):
                self.tokens_per_sec.set(float(m.tokens_per_sec))
            if getattr(m, "step", None) is not None:
                self.step.set(int(m.step))
                self.steps_total.inc()
            if getattr(m, "epoch", None) is not None:
                self.epoch.set(int(m.epoch))
            imb = getattr(m, "expert_imbalance", None)
            if imb is None:
                es = getattr(m, "expert_stats", None)
                if isinstance(es, dict):
                    imb = es.get("imbalance")
            if imb is not None:
                self.expert_imbalance.set(float(imb))
            mem = getattr(m, "memory_gb", None)
            if mem is None:
                mem = getattr(m, "memory_allocated_gb", None)
            if mem is not None:
                self.memory_gb.set(float(mem))
        finally:
            if self.next_hook is not None:
                self.next_hook(m)

    def scrape(self) -> bytes:
        return generate_latest(self.registry)


def make_server_metrics(registry=None):
    """Counters/histograms for the HTTP inference server."""
    if not HAVE_PROM:
        return None
    registry = registry or CollectorRegistry()
    return {
        "registry": registry,
        "requests": Counter("lumina_serve_requests_total",
                            "completion requests", ["endpoint"],
                            registry=registry),
        "tokens": Counter("lumina_serve_tokens_total",
                          "generated tokens", registry=registry),
        "latency": Histogram("lumina_serve_latency_seconds",
                             "end-to-end request latency",
                             registry=registry),
    }
