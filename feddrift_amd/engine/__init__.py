from .fljob import FLJob, run_iteration
from .timeline import run_timeline

__all__ = ["FLJob", "run_iteration", "run_timeline"]
