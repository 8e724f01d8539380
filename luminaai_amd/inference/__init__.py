from .engine import GenerationConfig, GenerationEngine, SAMPLING_MODES
from .loader import (find_latest_checkpoint, infer_config_from_state_dict,
                     load_checkpoint_smart, load_zero_shards)
from .chat import ChatInterface

__all__ = [
    "ChatInterface", "GenerationConfig", "GenerationEngine", "SAMPLING_MODES",
    "find_latest_checkpoint", "infer_config_from_state_dict",
    "load_checkpoint_smart", "load_zero_shards",
]
