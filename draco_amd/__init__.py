"""draco_amd — MI355X-native Byzantine-resilient distributed training.

A from-scratch rebuild of the capabilities of hwang595/Draco (ICML 2018,
arXiv:1803.09877) designed for AMD Instinct MI355X (gfx950):
PyTorch-ROCm model compute, hand-written HIP/CDNA4 kernels for every
coding/aggregation/update hot spot, RCCL over xGMI for all communication.
"""
__version__ = "0.1.0"

from .config import Config, parse_cli

__all__ = ["Config", "parse_cli", "__version__"]
