"""Stochastic-rounding ops (reference: d9d/kernel/stochastic)."""

import torch

from ._ext import get_ext


def _sr_cpu(src: torch.Tensor, seed: int) -> torch.Tensor:
    """Pure-torch stochastic rounding (test oracle)."""
    gen = torch.Generator(device="cpu").manual_seed(seed)
    bits = src.float().view(torch.int32).to(torch.int64) & 0xFFFFFFFF
    noise = torch.randint(0, 1 << 16, src.shape, generator=gen, dtype=torch.int64)
    nan_mask = torch.isnan(src)
    rounded = ((bits + noise) >> 16).to(torch.int32) << 16
    out = rounded.view(torch.float32).to(torch.bfloat16)
    out[nan_mask] = float("nan")
    return out


def copy_fp32_to_bf16_stochastic_(dst: torch.Tensor, src: torch.Tensor, seed: int) -> None:
    """dst (bf16) <- stochastic_round(src (fp32))."""
    if src.is_cuda:
        get_ext().copy_fp32_to_bf16_stochastic_(dst, src.contiguous(), seed)
        return
    dst.copy_(_sr_cpu(src, seed))


def adamw_stochastic_bf16_(
    param: torch.Tensor,
    grad: torch.Tensor,
    exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor,
    *,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    step: int,
    seed: int,
) -> None:
    """Fused AdamW with bf16 params/grads, fp32 moments and SR bf16 writes."""
    if param.is_cuda:
        get_ext().adamw_stochastic_bf16_(
            param, grad.contiguous(), exp_avg, exp_avg_sq,
            lr, beta1, beta2, eps, weight_decay, step, seed,
        )
        return
    # CPU reference path: fp32 math, RNE writes (SR is tested statistically on GPU).
    p32 = param.float()
    g32 = grad.float()
    p32.mul_(1 - lr * weight_decay)
    exp_avg.mul_(beta1).add_(g32, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g32, g32, value=1 - beta2)
    bc1 = 1 - beta1**step
    bc2 = 1 - beta2**step
    denom = (exp_avg_sq / bc2).sqrt_().add_(eps)
    p32.addcdiv_(exp_avg / bc1, denom, value=-lr)
    param.copy_(_sr_cpu(p32, seed))
