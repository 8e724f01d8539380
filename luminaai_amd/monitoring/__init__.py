from .logger import MetricsCollector, TrainingHealthMonitor, ProductionLogger

__all__ = ["MetricsCollector", "TrainingHealthMonitor", "ProductionLogger"]
