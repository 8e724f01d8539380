"""LuminaAI-AMD: an MI355X-native sparse-transformer training framework.

Brand-new implementation of the capabilities of MatN23/LuminaAI ("Adaptive
Training System") designed for AMD Instinct MI355X (gfx950, CDNA4):
PyTorch-ROCm autograd shell, hand-written HIP kernels (MFMA/LDS-tiled) for the
hot ops, RCCL collectives over xGMI for DP/ZeRO/EP parallelism.
"""

__version__ = "0.1.0"

from .config import Config, ConfigManager, ConfigPresets

__all__ = ["Config", "ConfigManager", "ConfigPresets", "__version__"]
