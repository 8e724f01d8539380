from .checkpoint import CheckpointManager
from .optimizer import FlatAdamW, split_decay_groups
from .precision import PrecisionManager
from .schedulers import WarmupScheduler, create_scheduler
from .trainer import EnhancedConversationTrainer, Trainer, TrainingMetrics

__all__ = [
    "CheckpointManager", "EnhancedConversationTrainer", "FlatAdamW",
    "PrecisionManager", "Trainer", "TrainingMetrics", "WarmupScheduler",
    "create_scheduler", "split_decay_groups",
]
