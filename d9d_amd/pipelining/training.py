"""Multi-stage optimizer/scheduler aggregates (reference: d9d/pipelining/training/).

A rank running several pipeline stages can either hand all stage parameters
to one optimizer (what the Trainer does) or keep one optimizer per stage and
drive them together through these aggregates — the reference's
PipelinedOptimizer/PipelinedLRScheduler API.
"""

from typing import Any

import torch

from ..core.offload import SleepTag, offload_tensor, onload_tensor


class PipelinedOptimizer:
    def __init__(self, optimizers: list[torch.optim.Optimizer]) -> None:
        self.optimizers = list(optimizers)

    @property
    def param_groups(self):
        return [g for opt in self.optimizers for g in opt.param_groups]

    @property
    def parameters(self):
        for opt in self.optimizers:
            for group in opt.param_groups:
                yield from group["params"]

    def step(self) -> None:
        for opt in self.optimizers:
            opt.step()

    def zero_grad(self, set_to_none: bool = True) -> None:
        for opt in self.optimizers:
            opt.zero_grad(set_to_none=set_to_none)

    def state_dict(self) -> dict[str, Any]:
        return {f"stage_{i}": opt.state_dict() for i, opt in enumerate(self.optimizers)}

    def load_state_dict(self, state: dict[str, Any]) -> None:
        for i, opt in enumerate(self.optimizers):
            key = f"stage_{i}"
            if key in state:
                opt.load_state_dict(state[key])

    # Offloadable (reference: training/optimizer.py Offloadable)
    def offload(self, tags) -> None:
        if SleepTag.OPTIMIZER in tags:
            for opt in self.optimizers:
                for st in opt.state.values():
                    for v in st.values():
                        if isinstance(v, torch.Tensor):
                            offload_tensor(v)

    def onload(self, tags) -> None:
        if SleepTag.OPTIMIZER in tags:
            device = (
                torch.device("cuda", torch.cuda.current_device())
                if torch.cuda.is_available()
                else torch.device("cpu")
            )
            for opt in self.optimizers:
                for st in opt.state.values():
                    for v in st.values():
                        if isinstance(v, torch.Tensor):
                            onload_tensor(v, device)


class PipelinedLRScheduler:
    def __init__(self, schedulers: list) -> None:
        self.schedulers = list(schedulers)

    def step(self) -> None:
        for s in self.schedulers:
            s.step()

    def get_last_lr(self) -> list[float]:
        out: list[float] = []
        for s in self.schedulers:
            out.extend(s.get_last_lr())
        return out

    def state_dict(self) -> dict[str, Any]:
        return {f"stage_{i}": s.state_dict() for i, s in enumerate(self.schedulers)}

    def load_state_dict(self, state: dict[str, Any]) -> None:
        for i, s in enumerate(self.schedulers):
            key = f"stage_{i}"
            if key in state:
                s.load_state_dict(state[key])
