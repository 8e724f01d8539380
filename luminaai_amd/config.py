"""Configuration system for the LuminaAI-AMD framework.

API-compatible re-design of the reference configuration layer
(reference: Src/Main_Scripts/config/config_manager.py:15-158 `Config`,
:759-1669 `ConfigPresets`, :1871 `ConfigManager`), built MI355X-first:

- hardware auto-configuration probes ROCm (gfx950) instead of CUDA compute
  capability: bf16 is always the training dtype on MI355X, fp8 (OCP e4m3fn)
  is available for GEMM paths;
- auto ZeRO staging is sized for 288 GB HBM3E per GPU (a ~70B-total model
  still fits un-sharded optimizer state at stage 2 on an 8-GPU node);
- communication defaults (bucket sizes) are tuned for RCCL over xGMI
  (7 point-to-point links x ~153 GB/s per GPU) rather than NVSwitch:
  several in-flight ~100 MB buckets instead of single 1 GB buckets.
"""

from __future__ import annotations

import dataclasses
import json
import math
import os
from dataclasses import dataclass, field
from typing import Any, Dict, Optional

try:
    import yaml
except ImportError:  # pragma: no cover
    yaml = None


@dataclass
class Config:
    """Training configuration. Field names match the reference `Config`
    (config_manager.py:15-158) so presets / YAML files / user code carry over."""

    # Model architecture
    vocab_size: int = 50304
    hidden_size: int = 512
    num_layers: int = 8
    num_heads: int = 8
    num_kv_heads: int = 4
    seq_length: int = 1024
    intermediate_size: Optional[int] = None  # auto: 8/3 * hidden, rounded to 256
    rms_norm_eps: float = 1e-6
    rope_theta: float = 10000.0
    dropout: float = 0.0

    # Training parameters
    batch_size: int = 2
    micro_batch_size: Optional[int] = None
    gradient_accumulation_steps: int = 8
    learning_rate: float = 1e-4
    weight_decay: float = 0.01
    num_epochs: int = 3
    warmup_ratio: float = 0.15
    eval_every_n_batches: int = 500
    save_every_n_batches: int = 1000
    precision: str = "auto"
    inference_precision: str = "auto"
    compile: bool = False  # hipGraphs + hand-written kernels instead of a tracing compiler

    # Data parameters
    train_data_path: str = "data/train.jsonl"
    eval_data_path: str = "data/eval.jsonl"
    num_workers: int = 2
    assistant_loss_weight: float = 1.5
    max_conversations_per_file: int = 10000
    streaming_threshold_gb: float = 10.0
    prefetch_factor: int = 4
    pin_memory: bool = True

    # Generation parameters
    max_new_tokens: int = 512
    temperature: float = 0.8
    top_p: float = 0.9
    top_k: int = 50

    # Stability and optimization
    init_std: float = 0.02
    layer_norm_eps: float = 1e-5
    use_stable_embedding: bool = True
    gradient_checkpointing: bool = True
    tie_word_embeddings: bool = True
    use_flash_attention: bool = True

    # MoE parameters
    use_moe: bool = True
    use_mod: bool = True
    num_experts: int = 8
    moe_top_k: int = 1
    capacity_factor: float = 1.5
    load_balancing_weight: float = 0.001
    expert_parallel_size: Optional[int] = None
    routing_temperature: float = 1.0
    routing_noise_std: float = 0.1
    moe_pattern: str = "all"
    mod_capacity_factor: float = 0.5
    mod_routing_temperature: float = 1.0

    # Distributed engine parameters (native ZeRO engine replaces
    # DeepSpeed/FSDP/ColossalAI; flag names kept for config compatibility)
    use_deepspeed: bool = False
    zero_stage: int = 0
    cpu_offload: bool = False
    fp8_alltoall: bool = False   # e4m3/e5m2 EP token exchange (halves a2a bytes)
    cpu_offload_optimizer: bool = False
    cpu_offload_parameters: bool = False
    aggressive_cpu_offload: bool = False
    nvme_path: Optional[str] = None
    nvme_offload_optimizer: bool = False
    nvme_offload_parameters: bool = False
    gradient_compression: bool = False
    communication_backend: str = "nccl"  # "nccl" == RCCL on ROCm
    overlap_comm: bool = True
    contiguous_gradients: bool = True
    # xGMI is 7 point-to-point links/GPU: several in-flight ~100 MB buckets
    # beat one 1 GB bucket (reference used 5e8 elements ~ 1 GB).
    allgather_bucket_size: int = 50_000_000
    reduce_bucket_size: int = 50_000_000

    # Production settings
    experiment_name: Optional[str] = None
    seed: int = 42
    log_level: str = "INFO"
    save_total_limit: int = 5
    early_stopping_patience: Optional[int] = None
    min_lr: float = 1e-6
    lr_scheduler: str = "cosine"
    use_lr_scheduler: bool = True

    # Monitoring and fault tolerance
    health_check_interval: int = 100
    # orchestrator metric emission cadence (optimizer steps). Emission reads
    # device scalars (loss) -> one sync per emission; at sub-10ms steps that
    # plus the monitor thread's GIL share costs ~35%, at ~1s steps <0.5%.
    # Raise for small/fast models.
    metrics_emit_every: int = 1
    auto_resume: bool = True
    backup_every_n_hours: int = 6
    max_retries: int = 3
    enable_wandb: bool = False
    prometheus_port: Optional[int] = None   # rank-0 scrape endpoint
    wandb_project: Optional[str] = None
    wandb_entity: Optional[str] = None

    # Adaptive learning-rate control
    enable_adaptive_lr: bool = True
    allow_scheduler_override: bool = True
    min_override_threshold: float = 0.2
    emergency_override_enabled: bool = True
    log_lr_decisions: bool = True

    # Advanced precision settings
    auto_tune_precision: bool = True
    precision_target: str = "balanced"
    dynamic_precision: bool = False
    tf32_enabled: Optional[bool] = None  # no xf32/TF32 on gfx950; kept for compat
    fp16_loss_scale: float = 65536.0
    bf16_enabled: bool = True
    fp8_enabled: bool = False  # OCP e4m3fn GEMM paths on gfx950

    # Memory optimization
    max_memory_usage: float = 0.9
    memory_cleanup_interval: int = 1000
    enable_cpu_adam: bool = False
    partition_activations: bool = False

    # Multi-node settings
    master_addr: Optional[str] = None
    master_port: int = 29500
    world_size: Optional[int] = None
    rank: Optional[int] = None
    local_rank: Optional[int] = None

    # Data processing
    data_cache_dir: str = "data/cache"
    tokenizer_cache_dir: str = "tokenizers/cache"
    max_seq_length_percentile: float = 0.95

    # Checkpointing enhancements
    save_optimizer_states: bool = True
    checkpoint_compression: bool = True
    async_save: bool = True
    universal_checkpoint: bool = True

    # Performance profiling
    profile_memory: bool = False
    profile_communication: bool = False
    log_throughput: bool = True

    # internal
    _batch_size_set: bool = field(default=False, init=False, repr=False)
    _device_optimizations_applied: bool = field(default=False, init=False, repr=False)

    # ------------------------------------------------------------------
    def __post_init__(self):
        self._auto_configure()
        self.validate()

    def _auto_configure(self):
        """Hardware- and arch-derived defaults (reference config_manager.py:164-313,
        re-done for ROCm/MI355X)."""
        # vocab to a multiple of 128 (matches tokenizer padding; MFMA-friendly)
        if self.vocab_size % 128 != 0:
            self.vocab_size = ((self.vocab_size + 127) // 128) * 128

        if self.intermediate_size is None:
            # SwiGLU sizing: 8/3 * hidden rounded up to a multiple of 256
            self.intermediate_size = ((int(self.hidden_size * 8 / 3) + 255) // 256) * 256

        if self.micro_batch_size is None:
            self.micro_batch_size = max(1, self.batch_size // max(1, self.gradient_accumulation_steps))

        if self.expert_parallel_size is None and self.use_moe:
            ws = self.world_size or _env_world_size() or 1
            self.expert_parallel_size = largest_divisor_leq(ws, self.num_experts)

        # auto precision: MI355X is bf16-first; fp32 on CPU
        if self.precision == "auto":
            self.precision = "bf16" if _rocm_available() else "fp32"
        if self.inference_precision == "auto":
            self.inference_precision = self.precision

        # auto ZeRO stage by total parameter count, sized for 288 GB HBM3E/GPU.
        # fp32 master + AdamW m,v + bf16 weights + bf16 grads = 18 bytes/param:
        # stage 0 to ~10B, stage 1 to ~40B, stage 2 to ~120B, stage 3 beyond.
        if self.zero_stage == 0 and self.use_deepspeed:
            p = self.estimate_total_params()
            if p > 120e9:
                self.zero_stage = 3
            elif p > 40e9:
                self.zero_stage = 2
            elif p > 10e9:
                self.zero_stage = 1

    # ------------------------------------------------------------------
    def _n_moe_layers(self) -> int:
        """Layers that are actually MoE under moe_pattern (models/
        transformer.py moe_layer_selector)."""
        if not self.use_moe:
            return 0
        from .models.transformer import moe_layer_selector
        return sum(moe_layer_selector(i, self.num_layers,
                                      getattr(self, "moe_pattern", "all"))
                   for i in range(self.num_layers))

    def estimate_active_params(self) -> int:
        h, L, V = self.hidden_size, self.num_layers, self.vocab_size
        inter = self.intermediate_size
        kv = self.num_kv_heads or self.num_heads
        head_dim = h // self.num_heads
        attn = h * h + 2 * h * kv * head_dim + h * h  # q,k,v,o
        ffn = 3 * h * inter  # gate, up, down
        n_moe = self._n_moe_layers()
        body = L * attn + n_moe * ffn * self.moe_top_k             + (L - n_moe) * ffn
        emb = V * h * (1 if self.tie_word_embeddings else 2)
        return int(body + emb)

    def estimate_total_params(self) -> int:
        h, L, V = self.hidden_size, self.num_layers, self.vocab_size
        inter = self.intermediate_size
        kv = self.num_kv_heads or self.num_heads
        head_dim = h // self.num_heads
        attn = h * h + 2 * h * kv * head_dim + h * h
        ffn = 3 * h * inter
        n_moe = self._n_moe_layers()
        body = L * attn + n_moe * (ffn * self.num_experts + h * self.num_experts)             + (L - n_moe) * ffn
        emb = V * h * (1 if self.tie_word_embeddings else 2)
        return int(body + emb)

    def estimate_memory_gb(self) -> float:
        """Per-GPU training memory estimate (weights+grads+optimizer, bf16 compute)."""
        p = self.estimate_total_params()
        bytes_per_param = 18.0  # bf16 w + bf16 g + fp32 master + fp32 m + fp32 v
        ws = self.world_size or 1
        if self.zero_stage >= 3:
            bytes_per_param = 18.0 / ws
        elif self.zero_stage == 2:
            bytes_per_param = 4.0 + 14.0 / ws
        elif self.zero_stage == 1:
            bytes_per_param = 6.0 + 12.0 / ws
        return p * bytes_per_param / 1e9

    # ------------------------------------------------------------------
    def validate(self):
        assert self.hidden_size % self.num_heads == 0, (
            f"hidden_size {self.hidden_size} not divisible by num_heads {self.num_heads}")
        assert self.num_heads % (self.num_kv_heads or self.num_heads) == 0, (
            f"num_heads {self.num_heads} not divisible by num_kv_heads {self.num_kv_heads}")
        assert self.moe_top_k <= self.num_experts
        assert 0.0 < self.mod_capacity_factor <= 1.0
        assert self.precision in ("auto", "fp32", "fp16", "bf16", "fp8", "mixed_bf16", "mixed_fp16")
        assert self.zero_stage in (0, 1, 2, 3)
        return True

    # ------------------------------------------------------------------
    def to_dict(self) -> Dict[str, Any]:
        d = dataclasses.asdict(self)
        d.pop("_batch_size_set", None)
        d.pop("_device_optimizations_applied", None)
        return d

    def save(self, path: str):
        d = self.to_dict()
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        with open(path, "w") as f:
            if yaml is not None and (path.endswith(".yaml") or path.endswith(".yml")):
                yaml.safe_dump(d, f, sort_keys=False)
            else:
                json.dump(d, f, indent=2)

    @classmethod
    def load(cls, path: str) -> "Config":
        with open(path) as f:
            if yaml is not None and (path.endswith(".yaml") or path.endswith(".yml")):
                d = yaml.safe_load(f)
            else:
                d = json.load(f)
        known = {f.name for f in dataclasses.fields(cls) if f.init}
        return cls(**{k: v for k, v in d.items() if k in known})


# ----------------------------------------------------------------------
def _rocm_available() -> bool:
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False


def _env_world_size() -> Optional[int]:
    v = os.environ.get("WORLD_SIZE")
    return int(v) if v else None


def largest_divisor_leq(n: int, cap: int) -> int:
    """Largest divisor of n that is <= cap (expert-parallel sizing;
    reference trainer.py:890-917 divisor scoring)."""
    best = 1
    for d in range(1, n + 1):
        if n % d == 0 and d <= cap:
            best = d
    return best


# ----------------------------------------------------------------------
class ConfigPresets:
    """Model-size presets. Architecture shapes match the reference presets
    (config_manager.py:759-1669) so benchmark configs line up 1:1."""

    @staticmethod
    def debug() -> Config:
        return Config(
            vocab_size=1024, hidden_size=128, num_layers=2, num_heads=2,
            num_kv_heads=1, seq_length=256, intermediate_size=256,
            batch_size=2, micro_batch_size=1, gradient_accumulation_steps=2,
            num_epochs=1, learning_rate=5e-5, weight_decay=0.01,
            eval_every_n_batches=50, save_every_n_batches=100,
            num_workers=0,
            use_moe=True, use_mod=False, num_experts=32, moe_top_k=2,
            capacity_factor=1.1, load_balancing_weight=0.005,
            expert_parallel_size=2,
            zero_stage=1,
            experiment_name="debug_run", log_level="DEBUG",
            health_check_interval=10, save_total_limit=3, max_retries=1,
            gradient_checkpointing=False,
            max_memory_usage=0.7, streaming_threshold_gb=1.0,
        )

    @staticmethod
    def debug_300m() -> Config:
        return Config(
            # 768, not the reference's 786 (config_manager.py:823) — 786 is
            # not divisible by num_heads=4 and would break head_dim
            vocab_size=50304, hidden_size=768, num_layers=6, num_heads=4,
            num_kv_heads=4, seq_length=512,
            batch_size=4, micro_batch_size=1, gradient_accumulation_steps=4,
            use_moe=True, use_mod=False, num_experts=8, moe_top_k=2,
            zero_stage=1, experiment_name="debug_300m",
        )

    @staticmethod
    def moe_stress_test() -> Config:
        return Config(
            vocab_size=50304, hidden_size=768, num_layers=6, num_heads=8,
            num_kv_heads=2, seq_length=256,
            batch_size=8, micro_batch_size=2, gradient_accumulation_steps=4,
            use_moe=True, use_mod=False, num_experts=32, moe_top_k=2,
            capacity_factor=1.25, load_balancing_weight=0.01,
            zero_stage=0, experiment_name="moe_stress_test",
        )

    @staticmethod
    def debug_200m() -> Config:
        return Config(
            vocab_size=50304, hidden_size=640, num_layers=12, num_heads=8,
            num_kv_heads=8, seq_length=512,
            batch_size=4, micro_batch_size=1, gradient_accumulation_steps=4,
            use_moe=False, use_mod=True, mod_capacity_factor=0.5,
            num_experts=32, moe_top_k=2,
            zero_stage=1, experiment_name="debug_200m",
        )

    @staticmethod
    def b1() -> Config:
        """~1B active; with use_moe=True this is the 8x1B ~ 8B-total headline config."""
        return Config(
            hidden_size=1908, num_layers=31, num_heads=12, num_kv_heads=4,
            seq_length=2048,
            batch_size=8, micro_batch_size=1, gradient_accumulation_steps=4,
            num_epochs=3, learning_rate=3e-4, weight_decay=0.01,
            use_moe=False, use_mod=True, mod_capacity_factor=0.6,
            num_experts=8, moe_top_k=1,
            capacity_factor=1.25, load_balancing_weight=0.01,
            zero_stage=2, experiment_name="b1_8x1b",
            early_stopping_patience=5, gradient_checkpointing=True,
            max_memory_usage=0.85, streaming_threshold_gb=5.0,
        )

    @staticmethod
    def b1_moe() -> Config:
        """The headline benchmark config: 8-expert top-2 MoE, ~1.3B active / ~8B total."""
        c = ConfigPresets.b1()
        c.use_moe = True
        c.use_mod = False
        c.num_experts = 8
        c.moe_top_k = 2
        c.experiment_name = "b1_moe_8e_top2"
        return c

    @staticmethod
    def b7() -> Config:
        return Config(
            hidden_size=4096, num_layers=32, num_heads=32, num_kv_heads=8,
            seq_length=4096,
            batch_size=16, micro_batch_size=1, gradient_accumulation_steps=8,
            learning_rate=1e-4,
            use_moe=False, use_mod=True, mod_capacity_factor=0.5,
            num_experts=8, moe_top_k=1, capacity_factor=1.25,
            load_balancing_weight=0.01,
            zero_stage=0, experiment_name="b7_8x7b_mixtral",
            gradient_checkpointing=True,
        )

    @staticmethod
    def b7_moe() -> Config:
        c = ConfigPresets.b7()
        c.use_moe = True
        c.use_mod = False
        c.moe_top_k = 2
        c.zero_stage = 2
        c.experiment_name = "b7_moe_8e_top2"
        return c

    @staticmethod
    def b14() -> Config:
        return Config(
            hidden_size=5120, num_layers=40, num_heads=40, num_kv_heads=10,
            seq_length=4096, batch_size=16, gradient_accumulation_steps=8,
            use_moe=False, use_mod=True, num_experts=8, moe_top_k=1,
            zero_stage=1, experiment_name="b14", gradient_checkpointing=True,
        )

    @staticmethod
    def b30() -> Config:
        return Config(
            hidden_size=6656, num_layers=48, num_heads=52, num_kv_heads=13,
            seq_length=8192, batch_size=16, gradient_accumulation_steps=16,
            use_moe=False, use_mod=True, num_experts=8, moe_top_k=1,
            zero_stage=2, experiment_name="b30", gradient_checkpointing=True,
        )

    @staticmethod
    def b50() -> Config:
        return Config(
            hidden_size=8192, num_layers=56, num_heads=64, num_kv_heads=16,
            seq_length=8192, batch_size=16, gradient_accumulation_steps=16,
            use_moe=False, use_mod=True, num_experts=8, moe_top_k=1,
            zero_stage=3, experiment_name="b50", gradient_checkpointing=True,
        )

    @staticmethod
    def b75() -> Config:
        return Config(
            hidden_size=10240, num_layers=64, num_heads=80, num_kv_heads=20,
            seq_length=8192, batch_size=16, gradient_accumulation_steps=32,
            use_moe=False, use_mod=True, num_experts=8, moe_top_k=1,
            zero_stage=3, experiment_name="b75", gradient_checkpointing=True,
        )

    @staticmethod
    def b100() -> Config:
        return Config(
            hidden_size=12288, num_layers=72, num_heads=96, num_kv_heads=24,
            seq_length=8192, batch_size=16, gradient_accumulation_steps=32,
            use_moe=False, use_mod=True, num_experts=8, moe_top_k=1,
            zero_stage=3, experiment_name="b100", gradient_checkpointing=True,
        )

    @staticmethod
    def b200() -> Config:
        return Config(
            hidden_size=16384, num_layers=88, num_heads=128, num_kv_heads=32,
            seq_length=8192, batch_size=16, gradient_accumulation_steps=64,
            use_moe=False, use_mod=True, num_experts=8, moe_top_k=1,
            zero_stage=3, experiment_name="b200", gradient_checkpointing=True,
        )

    @staticmethod
    def b300() -> Config:
        return Config(
            hidden_size=20480, num_layers=96, num_heads=160, num_kv_heads=40,
            seq_length=8192, batch_size=16, gradient_accumulation_steps=64,
            use_moe=False, use_mod=True, num_experts=8, moe_top_k=1,
            zero_stage=3, experiment_name="b300", gradient_checkpointing=True,
        )

    @staticmethod
    def hybrid_70b() -> Config:
        """Hybrid MoE(64-expert top-2)+MoD ~70B-total (BASELINE.json config #5)."""
        return Config(
            hidden_size=2048, num_layers=32, num_heads=16, num_kv_heads=4,
            seq_length=2048, batch_size=8, gradient_accumulation_steps=4,
            use_moe=True, use_mod=True, num_experts=64, moe_top_k=2,
            mod_capacity_factor=0.5, capacity_factor=1.25,
            load_balancing_weight=0.01, moe_pattern="every_2nd",
            zero_stage=3, fp8_enabled=True,
            experiment_name="hybrid_70b_fp8", gradient_checkpointing=True,
        )

    @staticmethod
    def names():
        return [n for n in dir(ConfigPresets)
                if not n.startswith("_") and n not in ("names", "get")]

    @staticmethod
    def get(name: str) -> Config:
        fn = getattr(ConfigPresets, name, None)
        if fn is None:
            raise KeyError(f"unknown preset '{name}'; have {ConfigPresets.names()}")
        return fn()


# ----------------------------------------------------------------------
class ConfigManager:
    """Load/save/validate configs + hardware optimization entry point
    (reference config_manager.py:1871-2099)."""

    @staticmethod
    def load_config(path: str) -> Config:
        return Config.load(path)

    @staticmethod
    def save_config(config: Config, path: str):
        config.save(path)

    @staticmethod
    def validate_config(config: Config) -> bool:
        return config.validate()

    @staticmethod
    def optimize_for_hardware(config: Config) -> Config:
        """Apply MI355X-specific tuning: bf16 precision, bucket sizes for xGMI,
        micro-batch fitting against 288 GB HBM3E."""
        if _rocm_available():
            config.precision = "bf16" if config.precision in ("auto", "fp32") else config.precision
            import torch
            total = torch.cuda.get_device_properties(0).total_memory
            budget = total * config.max_memory_usage
            est = config.estimate_memory_gb() * 1e9
            if est > budget and config.zero_stage < 2:
                config.zero_stage = 2
        config._device_optimizations_applied = True
        return config
