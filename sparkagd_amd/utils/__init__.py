from .checkpoint import save_checkpoint, load_checkpoint
from .metrics import JsonlMetrics, NullMetrics
from .profiling import profile_run

__all__ = ["save_checkpoint", "load_checkpoint", "JsonlMetrics", "NullMetrics", "profile_run"]
