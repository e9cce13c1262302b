from .profiling import trace_range, StepTimer
from .checkpoint import save_checkpoint, load_checkpoint

__all__ = ["trace_range", "StepTimer", "save_checkpoint", "load_checkpoint"]
