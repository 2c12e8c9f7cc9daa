from .metrics import MetricLogger

__all__ = ["MetricLogger"]
