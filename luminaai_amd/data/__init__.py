from .tokenizer import ConversationTokenizer
from .dataset import (
    BaseTrainingDataset,
    ConversationDataset,
    HybridDatasetManager,
    InterleavedDataset,
    StreamingBaseTrainingDataset,
    SyntheticDataset,
    create_dataloader,
    setup_datasets,
)

__all__ = [
    "BaseTrainingDataset", "ConversationDataset", "ConversationTokenizer",
    "HybridDatasetManager", "InterleavedDataset", "StreamingBaseTrainingDataset",
    "SyntheticDataset", "create_dataloader", "setup_datasets",
]
