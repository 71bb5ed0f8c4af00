from .checkpoint import load_checkpoint, save_checkpoint
from .logging import MetricsLogger

__all__ = ["MetricsLogger", "save_checkpoint", "load_checkpoint"]
