from .data_processing import (flatten_conversation_tree, generate_sample_data,
                              validate_jsonl)
from .environment import (estimate_training_time, get_system_info,
                          pick_device, validate_environment)
from .profiling import (get_profiling_stats, profile_function,
                        profiling_context, reset_profiling_stats)
from .reporting import create_training_report, data_summary

__all__ = [
    "create_training_report", "data_summary", "estimate_training_time",
    "flatten_conversation_tree", "generate_sample_data",
    "get_profiling_stats", "get_system_info", "pick_device",
    "profile_function", "profiling_context", "reset_profiling_stats",
    "validate_environment", "validate_jsonl",
]
