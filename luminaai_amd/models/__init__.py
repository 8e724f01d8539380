from .transformer import (
    DeepSeekConfig,
    DeepSeekTransformer,
    GroupedQueryAttention,
    KVCache,
    MoDRouter,
    MoEFFNLayer,
    RMSNorm,
    RotaryEmbedding,
    SwiGLUExpert,
    TransformerBlock,
    config_to_deepseek_config,
)

__all__ = [
    "DeepSeekConfig", "DeepSeekTransformer", "GroupedQueryAttention", "KVCache",
    "MoDRouter", "MoEFFNLayer", "RMSNorm", "RotaryEmbedding", "SwiGLUExpert",
    "TransformerBlock", "config_to_deepseek_config",
]
