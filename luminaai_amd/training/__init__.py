from .checkpoint import CheckpointManager
from .chinchilla import (AdaptiveCurriculumManager, ComputeEfficiencyTracker,
                         ConvergenceDetector, EnhancedChinchillaScaler)
from .optimizer import FlatAdamW, split_decay_groups
from .orchestrator import (AdaptiveDecision, AdaptiveTrainer,
                           AdaptiveTrainingOrchestrator, ArchitectureEvolution,
                           AdaptiveHyperparameterOptimizer, MetaLearningEngine,
                           ProductionMonitoring, RealTimeAnalytics)
from .precision import PrecisionManager
from .schedulers import WarmupScheduler, create_scheduler
from .trainer import EnhancedConversationTrainer, Trainer, TrainingMetrics

__all__ = [
    "AdaptiveCurriculumManager", "AdaptiveDecision",
    "AdaptiveHyperparameterOptimizer", "AdaptiveTrainer",
    "AdaptiveTrainingOrchestrator", "ArchitectureEvolution",
    "CheckpointManager", "ComputeEfficiencyTracker", "ConvergenceDetector",
    "EnhancedChinchillaScaler", "EnhancedConversationTrainer", "FlatAdamW",
    "MetaLearningEngine", "PrecisionManager", "ProductionMonitoring",
    "RealTimeAnalytics", "Trainer", "TrainingMetrics", "WarmupScheduler",
    "create_scheduler", "split_decay_groups",
]
