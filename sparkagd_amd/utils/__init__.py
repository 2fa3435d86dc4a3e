from .checkpoint import save_checkpoint, load_checkpoint
from .metrics import JsonlMetrics, NullMetrics

__all__ = ["save_checkpoint", "load_checkpoint", "JsonlMetrics", "NullMetrics"]
