from .logger import MetricsCollector, TrainingHealthMonitor, ProductionLogger
from .wandb_compat import WandbLogger

__all__ = ["MetricsCollector", "ProductionLogger", "TrainingHealthMonitor",
           "WandbLogger"]
