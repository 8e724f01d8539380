from .chat import ChatInterface
from .continuous import ContinuousBatchingEngine
from .engine import (SAMPLING_MODES, GenerationConfig, GenerationEngine,
                     sample_token)
from .loader import (find_ep_shards, find_latest_checkpoint, find_pp_stages,
                     infer_config_from_state_dict, load_checkpoint_smart,
                     load_zero_shards, merge_ep_checkpoints,
                     merge_pp_checkpoints)

__all__ = [
    "ChatInterface", "ContinuousBatchingEngine", "GenerationConfig",
    "GenerationEngine", "SAMPLING_MODES", "find_ep_shards",
    "find_latest_checkpoint", "find_pp_stages",
    "infer_config_from_state_dict", "load_checkpoint_smart",
    "load_zero_shards", "merge_ep_checkpoints", "merge_pp_checkpoints",
    "sample_token",
]
