from .training import OptimizerProtocol, LRSchedulerProtocol, Stateful

__all__ = ["OptimizerProtocol", "LRSchedulerProtocol", "Stateful"]
