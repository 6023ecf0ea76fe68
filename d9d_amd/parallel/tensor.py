"""Tensor parallelism (TP) + sequence parallelism (SP).

The reference DECLARES tp but rejects tp > 1
(d9d/module/parallelism/model/qwen3_moe.py:35-36); this is a real
implementation: colwise/rowwise DTensor sharding of attention and FFN
projections over the `tp` mesh dim with the boundary collectives on RCCL.

Forward math per block:
  x -> [copy-to-tp: fwd identity / bwd all-reduce]
    -> colwise projections (weight Shard(0), local out features)
    -> ... local-head attention or local-intermediate FFN ...
    -> rowwise projection (weight Shard(1), partial out)
    -> [fwd all-reduce / bwd identity] -> y

Sequence-parallel variant replaces the boundary collectives by
all-gather(seq) on entry and reduce-scatter(seq) on exit, so the
norm/residual region holds only S/tp of the sequence.
"""

import torch
import torch.distributed as dist
from torch import nn
from torch.distributed.device_mesh import DeviceMesh
from torch.distributed.tensor import DTensor, Placement, Replicate, Shard

from ..module.block.attention import GroupedQueryAttention
from ..module.block.ffn import SwiGLU
from .style import _to_local_class


class _CopyToTP(torch.autograd.Function):
    """fwd identity; bwd all-reduce (input used by all tp ranks)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):
        grad = grad.contiguous()
        dist.all_reduce(grad, group=ctx.group)
        return grad, None


class _SumGradFromTP(torch.autograd.Function):
    """fwd identity; bwd all-reduce — for params replicated inside the tp
    region (e.g. q/k norm weights) whose per-microbatch grads are partial."""

    @staticmethod
    def forward(ctx, w, group):
        ctx.group = group
        return w.view_as(w)

    @staticmethod
    def backward(ctx, grad):
        grad = grad.contiguous()
        dist.all_reduce(grad, group=ctx.group)
        return grad, None


_SUMGRAD_CLASS_CACHE: dict[tuple, type] = {}


def _install_tp_sumgrad(module: nn.Module, param_names: tuple[str, ...], group) -> None:
    """Swap `module`'s class so each named param reads through
    `_SumGradFromTP` during grad-enabled forward (identity otherwise)."""
    module._d9d_tp_group = group

    def _make_getter(name: str):
        def getter(self):
            p = self._parameters.get(name)
            if p is None:
                raise AttributeError(name)
            if torch.is_grad_enabled() and p.requires_grad:
                return _SumGradFromTP.apply(p, self._d9d_tp_group)
            return p

        return property(getter)

    key = (type(module), param_names)
    if key not in _SUMGRAD_CLASS_CACHE:
        ns = {name: _make_getter(name) for name in param_names}
        ns["_d9d_tp_sumgrad_params"] = param_names
        _SUMGRAD_CLASS_CACHE[key] = type(
            f"TPSumGrad{type(module).__name__}", (type(module),), ns
        )
    module.__class__ = _SUMGRAD_CLASS_CACHE[key]


class _ReduceFromTP(torch.autograd.Function):
    """fwd all-reduce (partial rowwise outputs); bwd identity."""

    @staticmethod
    def forward(ctx, x, group):
        x = x.contiguous()
        dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad, None


class _AllGatherSeq(torch.autograd.Function):
    """fwd all-gather along dim; bwd reduce-scatter (SP entry)."""

    @staticmethod
    def forward(ctx, x, group, dim):
        ctx.group = group
        ctx.dim = dim
        world = dist.get_world_size(group)
        parts = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(parts, x.contiguous(), group=group)
        return torch.cat(parts, dim=dim)

    @staticmethod
    def backward(ctx, grad):
        return _reduce_scatter_dim(grad, ctx.group, ctx.dim), None, None


def _reduce_scatter_dim(x: torch.Tensor, group, dim: int) -> torch.Tensor:
    """Sum across ranks, return this rank's chunk along `dim`.

    RCCL path: reduce_scatter_tensor (ring over xGMI). gloo's list-based
    reduce_scatter mis-reduces interleaved chunks, so the CPU path does
    all-reduce + slice.
    """
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    x = x.contiguous()
    if x.is_cuda and dim == 0:
        out = torch.empty_like(x.chunk(world, dim=0)[0])
        dist.reduce_scatter_tensor(out, x, group=group)
        return out
    if x.is_cuda:
        moved = x.movedim(dim, 0).contiguous()
        out = torch.empty_like(moved.chunk(world, dim=0)[0])
        dist.reduce_scatter_tensor(out, moved, group=group)
        return out.movedim(0, dim).contiguous()
    dist.all_reduce(x, group=group)
    return x.chunk(world, dim=dim)[rank].contiguous()


class _ReduceScatterSeq(torch.autograd.Function):
    """fwd reduce-scatter along dim; bwd all-gather (SP exit)."""

    @staticmethod
    def forward(ctx, x, group, dim):
        ctx.group = group
        ctx.dim = dim
        return _reduce_scatter_dim(x, group, dim)

    @staticmethod
    def backward(ctx, grad):
        world = dist.get_world_size(ctx.group)
        parts = [torch.empty_like(grad) for _ in range(world)]
        dist.all_gather(parts, grad.contiguous(), group=ctx.group)
        return torch.cat(parts, dim=ctx.dim), None, None


def _shard_linear(lin: nn.Linear, mesh: DeviceMesh, tp_dim: int, shard_dim: int) -> None:
    """Replace lin.weight with its LOCAL shard wrapped as a DTensor.

    shard_dim 0 = colwise (split out features); 1 = rowwise (split in features).
    """
    tp_size = mesh.shape[tp_dim]
    tp_rank = mesh.get_coordinate()[tp_dim]
    w = lin._parameters["weight"]
    full = w.data
    local = full.chunk(tp_size, dim=shard_dim)[tp_rank].contiguous()
    placements: list[Placement] = [Replicate()] * mesh.ndim
    placements[tp_dim] = Shard(shard_dim)
    dt = DTensor.from_local(local, mesh, tuple(placements), run_check=False)
    lin._parameters["weight"] = nn.Parameter(dt, requires_grad=w.requires_grad)
    if shard_dim == 0:
        lin.out_features = local.shape[0]
    else:
        lin.in_features = local.shape[1]
    if not getattr(lin, "_d9d_to_local_params", None):
        lin.__class__ = _to_local_class(type(lin), ("weight",))


def parallelize_tp_attention(
    attn: GroupedQueryAttention, mesh: DeviceMesh, tp_dim_name: str = "tp",
    sequence_parallel: bool = False,
) -> None:
    tp_dim = mesh.mesh_dim_names.index(tp_dim_name)
    tp_size = mesh.shape[tp_dim]
    if tp_size == 1:
        return
    group = mesh.get_group(tp_dim)
    assert attn.num_heads % tp_size == 0 and attn.num_kv_heads % tp_size == 0, (
        "attention heads must divide tp"
    )
    _shard_linear(attn.q_proj, mesh, tp_dim, 0)
    _shard_linear(attn.k_proj, mesh, tp_dim, 0)
    _shard_linear(attn.v_proj, mesh, tp_dim, 0)
    _shard_linear(attn.o_proj, mesh, tp_dim, 1)
    attn.num_heads //= tp_size
    attn.num_kv_heads //= tp_size

    # Replicated params INSIDE the tp region see only the local heads'
    # activations, so their gradients are partial: sum them over tp.
    # (q_norm/k_norm weights are shared across heads; learnable sinks are
    # per-head and follow the q-head shard instead.)
    #
    # The reduction lives IN the autograd graph (`_SumGradFromTP` wraps the
    # weight each forward), so each microbatch's partial grad is reduced
    # exactly once before it accumulates into .grad — a post-accumulate hook
    # would re-reduce earlier microbatches' already-summed contributions
    # under gradient accumulation.
    for norm in (attn.q_norm, attn.k_norm):
        if norm is not None:
            _install_tp_sumgrad(norm, ("weight",), group)
    if attn.sinks is not None:
        rank = dist.get_rank(group)
        shard = attn.sinks.shape[0] // tp_size
        with torch.no_grad():
            local_sinks = attn.sinks[rank * shard : (rank + 1) * shard].clone()
        attn.sinks = nn.Parameter(local_sinks)

    orig_forward = attn.forward

    def forward(hidden_states, rotary_cos_sin):
        if sequence_parallel:
            hidden_states = _AllGatherSeq.apply(hidden_states, group, 1)
        else:
            hidden_states = _CopyToTP.apply(hidden_states, group)
        out = orig_forward(hidden_states, rotary_cos_sin)
        if sequence_parallel:
            return _ReduceScatterSeq.apply(out, group, 1)
        return _ReduceFromTP.apply(out, group)

    attn.forward = forward


def parallelize_tp_mlp(
    mlp: SwiGLU, mesh: DeviceMesh, tp_dim_name: str = "tp",
    sequence_parallel: bool = False,
) -> None:
    tp_dim = mesh.mesh_dim_names.index(tp_dim_name)
    tp_size = mesh.shape[tp_dim]
    if tp_size == 1:
        return
    group = mesh.get_group(tp_dim)
    _shard_linear(mlp.gate_proj, mesh, tp_dim, 0)
    _shard_linear(mlp.up_proj, mesh, tp_dim, 0)
    _shard_linear(mlp.down_proj, mesh, tp_dim, 1)

    orig_forward = mlp.forward

    def forward(x):
        if sequence_parallel:
            x = _AllGatherSeq.apply(x, group, 1)
        else:
            x = _CopyToTP.apply(x, group)
        out = orig_forward(x)
        if sequence_parallel:
            return _ReduceScatterSeq.apply(out, group, 1)
        return _ReduceFromTP.apply(out, group)

    mlp.forward = forward


def parallelize_tensor_parallel(
    module: nn.Module,
    mesh: DeviceMesh,
    tp_dim_name: str = "tp",
    sequence_parallel: bool = False,
) -> nn.Module:
    """Apply TP (optionally SP) to every GQA attention and SwiGLU FFN block."""
    if mesh.mesh_dim_names and "pp" in mesh.mesh_dim_names and mesh.ndim > 1:
        # stages hold different modules; placements must not span pp
        mesh = mesh[tuple(n for n in mesh.mesh_dim_names if n != "pp")]
    for sub in module.modules():
        if isinstance(sub, GroupedQueryAttention):
            parallelize_tp_attention(sub, mesh, tp_dim_name, sequence_parallel)
        elif isinstance(sub, SwiGLU):
            parallelize_tp_mlp(sub, mesh, tp_dim_name, sequence_parallel)
    return module
