from .metrics import SmoothedValue, MetricLogger

__all__ = ["SmoothedValue", "MetricLogger"]
