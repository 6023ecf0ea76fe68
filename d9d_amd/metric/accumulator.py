"""Local/synced accumulator pair (reference: d9d/metric/component/accumulator.py:42)."""

import torch
import torch.distributed as dist


class MetricAccumulator:
    """Holds a local accumulator and a synced copy; sync = all-reduce."""

    def __init__(self, shape=(), dtype=torch.float64, op: str = "sum") -> None:
        self.local = torch.zeros(shape, dtype=dtype)
        self.synced = torch.zeros(shape, dtype=dtype)
        self.op = op

    def to(self, device):
        self.local = self.local.to(device)
        self.synced = self.synced.to(device)
        return self

    def add_(self, value) -> None:
        if self.op == "sum":
            self.local += value
        elif self.op == "max":
            self.local = torch.maximum(self.local, torch.as_tensor(value, dtype=self.local.dtype))
        elif self.op == "min":
            self.local = torch.minimum(self.local, torch.as_tensor(value, dtype=self.local.dtype))

    def sync(self, group=None) -> None:
        self.synced = self.local.clone()
        if dist.is_initialized():
            red = {"sum": dist.ReduceOp.SUM, "max": dist.ReduceOp.MAX, "min": dist.ReduceOp.MIN}[self.op]
            sync_t = self.synced
            if sync_t.dtype == torch.float64 and sync_t.is_cuda:
                sync_t = sync_t.to(torch.float32)
            dist.all_reduce(sync_t, op=red, group=group)
            self.synced = sync_t.to(self.synced.dtype)

    def reset(self) -> None:
        self.local.zero_()
