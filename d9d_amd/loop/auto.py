"""Config-driven optimizer/LR providers (reference: d9d/loop/auto/)."""

from typing import Literal

import torch
from pydantic import BaseModel

from ..lr_scheduler import CosineCurve, LinearCurve, Phase, PiecewiseLRScheduler
from .control import LRSchedulerProvider, OptimizerProvider


class OptimizerConfig(BaseModel):
    optimizer: Literal["stochastic_adamw", "adamw", "adam", "sgd"] = "adamw"
    lr: float = 1e-3
    betas: tuple[float, float] = (0.9, 0.95)
    eps: float = 1e-8
    weight_decay: float = 0.01
    momentum: float = 0.9  # sgd
    fused: bool = False


class AutoOptimizerProvider(OptimizerProvider):
    def __init__(self, config: OptimizerConfig) -> None:
        self.config = config

    def build_optimizer(self, named_params) -> torch.optim.Optimizer:
        cfg = self.config
        params = [p for _, p in named_params]
        if cfg.optimizer == "stochastic_adamw":
            from ..optim import StochasticAdamW

            return StochasticAdamW(
                params, lr=cfg.lr, betas=cfg.betas, eps=cfg.eps,
                weight_decay=cfg.weight_decay,
            )
        if cfg.optimizer == "adamw":
            return torch.optim.AdamW(
                params, lr=cfg.lr, betas=cfg.betas, eps=cfg.eps,
                weight_decay=cfg.weight_decay, fused=cfg.fused or None,
            )
        if cfg.optimizer == "adam":
            return torch.optim.Adam(
                params, lr=cfg.lr, betas=cfg.betas, eps=cfg.eps,
                fused=cfg.fused or None,
            )
        if cfg.optimizer == "sgd":
            return torch.optim.SGD(
                params, lr=cfg.lr, momentum=cfg.momentum,
                weight_decay=cfg.weight_decay,
            )
        raise ValueError(cfg.optimizer)


class LRSchedulerConfig(BaseModel):
    warmup_steps: int = 0
    decay_steps: int = 1000
    decay: Literal["cosine", "linear", "none"] = "cosine"
    min_lr_fraction: float = 0.0


class AutoLRSchedulerProvider(LRSchedulerProvider):
    def __init__(self, config: LRSchedulerConfig) -> None:
        self.config = config

    def build_lr_scheduler(self, optimizer):
        cfg = self.config
        phases = []
        if cfg.warmup_steps > 0:
            phases.append(Phase(cfg.warmup_steps, LinearCurve(0.0, 1.0)))
        if cfg.decay == "cosine":
            phases.append(Phase(cfg.decay_steps, CosineCurve(1.0, cfg.min_lr_fraction)))
        elif cfg.decay == "linear":
            phases.append(Phase(cfg.decay_steps, LinearCurve(1.0, cfg.min_lr_fraction)))
        else:
            phases.append(Phase(cfg.decay_steps, LinearCurve(1.0, 1.0)))
        return PiecewiseLRScheduler(optimizer, phases)
