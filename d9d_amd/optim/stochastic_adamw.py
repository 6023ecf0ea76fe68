"""StochasticAdamW — bf16 params, fp32 moments, stochastic-rounded writes.

Reference contract: d9d/optim/stochastic/adamw.py:40-195 — bf16 parameters
required, fp32 math, SR writes so repeated tiny updates do not vanish in
bf16, DTensor-aware (kernel runs on the local shard), own RNG state in
state_dict. Compute runs in the fused CDNA4 kernel (one launch per param).
"""

from typing import Any, Iterable

import torch
from torch.distributed.tensor import DTensor

from ..ops import adamw_stochastic_bf16_


def _local(t: torch.Tensor) -> torch.Tensor:
    return t.to_local() if isinstance(t, DTensor) else t


class StochasticAdamW(torch.optim.Optimizer):
    def __init__(
        self,
        params: Iterable,
        lr: float = 1e-3,
        betas: tuple[float, float] = (0.9, 0.95),
        eps: float = 1e-8,
        weight_decay: float = 0.01,
        seed: int = 0,
    ) -> None:
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self._seed = seed
        self._step_count = 0

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        self._step_count += 1
        idx = 0
        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            # GPU params with identical hyperparams batch into ONE multi-tensor
            # launch (the per-tensor kernel costs ~360 launches/step on the
            # bench model); CPU / odd params fall back to the single kernel.
            batch: list[tuple] = []
            for p in group["params"]:
                if p.grad is None:
                    continue
                local_p = _local(p.data)
                local_g = _local(p.grad)
                if local_p.dtype != torch.bfloat16:
                    raise TypeError(
                        "StochasticAdamW requires bf16 parameters, got "
                        f"{local_p.dtype} for a param of shape {tuple(p.shape)}"
                    )
                state = self.state[p]
                if not state:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(local_p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(local_p, dtype=torch.float32)
                state["step"] += 1
                idx += 1
                seed = (
                    self._seed * 0x9E3779B9
                    + self._step_count * 0x85EBCA6B
                    + idx * 0xC2B2AE35
                ) & 0x7FFFFFFFFFFFFFFF
                if local_p.is_cuda:
                    batch.append((
                        local_p.view(-1),
                        local_g.reshape(-1).to(torch.bfloat16).contiguous(),
                        state["exp_avg"].view(-1),
                        state["exp_avg_sq"].view(-1),
                        state["step"],
                        seed,
                    ))
                else:
                    adamw_stochastic_bf16_(
                        local_p.view(-1),
                        local_g.reshape(-1).to(torch.bfloat16),
                        state["exp_avg"].view(-1),
                        state["exp_avg_sq"].view(-1),
                        lr=group["lr"],
                        beta1=beta1,
                        beta2=beta2,
                        eps=group["eps"],
                        weight_decay=group["weight_decay"],
                        step=state["step"],
                        seed=seed,
                    )
            if batch:
                from ..ops._ext import get_ext

                get_ext().adamw_stochastic_bf16_multi_(
                    [b[0] for b in batch],
                    [b[1] for b in batch],
                    [b[2] for b in batch],
                    [b[3] for b in batch],
                    group["lr"], beta1, beta2, group["eps"],
                    group["weight_decay"],
                    [b[4] for b in batch],
                    [b[5] for b in batch],
                )
        return loss

    def state_dict(self) -> dict[str, Any]:
        sd = super().state_dict()
        sd["sr_rng"] = {"seed": self._seed, "step_count": self._step_count}
        return sd

    def load_state_dict(self, state_dict: dict[str, Any]) -> None:
        rng = state_dict.pop("sr_rng", None)
        super().load_state_dict(state_dict)
        if rng is not None:
            self._seed = rng["seed"]
            self._step_count = rng["step_count"]
