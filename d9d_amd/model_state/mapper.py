"""Model-state mapper DAG (reference: d9d/model_state/mapper/).

A mapper declares dependency groups (`StateGroup{inputs, outputs}`) and
transforms the input tensors of one group into its outputs. The streaming IO
layer fires each group as soon as all of its inputs are available, so a
full-model transform (e.g. HF -> d9d expert stacking) never materializes the
whole checkpoint in memory.
"""

from abc import ABC, abstractmethod
from dataclasses import dataclass

import torch
from torch import nn
from torch.distributed.device_mesh import DeviceMesh
from torch.distributed.tensor import DTensor, Placement, distribute_tensor


@dataclass(frozen=True)
class StateGroup:
    inputs: frozenset
    outputs: frozenset

    @staticmethod
    def of(inputs, outputs) -> "StateGroup":
        return StateGroup(frozenset(inputs), frozenset(outputs))


class ModelStateMapper(ABC):
    @abstractmethod
    def state_dependency_groups(self) -> list[StateGroup]: ...

    @abstractmethod
    def apply_group(self, group: StateGroup, tensors: dict) -> dict: ...

    def apply(self, state: dict) -> dict:
        """Run every group whose inputs are fully present."""
        out: dict = {}
        for group in self.state_dependency_groups():
            if all(k in state for k in group.inputs):
                out.update(self.apply_group(group, {k: state[k] for k in group.inputs}))
        return out


# ---- leaves -----------------------------------------------------------------


class _SingleTensor(ModelStateMapper):
    """One input key -> one output key with a tensor function."""

    def __init__(self, src: str, dst: str | None = None) -> None:
        self.src = src
        self.dst = dst if dst is not None else src

    def state_dependency_groups(self) -> list[StateGroup]:
        return [StateGroup.of([self.src], [self.dst])]

    def _fn(self, t: torch.Tensor) -> torch.Tensor:
        return t

    def apply_group(self, group, tensors):
        return {self.dst: self._fn(tensors[self.src])}


class Identity(_SingleTensor):
    pass


class Rename(_SingleTensor):
    def __init__(self, src: str, dst: str) -> None:
        super().__init__(src, dst)


class Transpose(_SingleTensor):
    def __init__(self, src: str, dst: str | None = None, dim0: int = 0, dim1: int = 1):
        super().__init__(src, dst)
        self.dim0, self.dim1 = dim0, dim1

    def _fn(self, t):
        return t.transpose(self.dim0, self.dim1).contiguous()


class Squeeze(_SingleTensor):
    def __init__(self, src: str, dst: str | None = None, dim: int = 0):
        super().__init__(src, dst)
        self.dim = dim

    def _fn(self, t):
        return t.squeeze(self.dim)


class Unsqueeze(_SingleTensor):
    def __init__(self, src: str, dst: str | None = None, dim: int = 0):
        super().__init__(src, dst)
        self.dim = dim

    def _fn(self, t):
        return t.unsqueeze(self.dim)


class CastDType(_SingleTensor):
    def __init__(self, src: str, dst: str | None = None, dtype: torch.dtype = torch.bfloat16):
        super().__init__(src, dst)
        self.dtype = dtype

    def _fn(self, t):
        return t.to(self.dtype)


class SelectChild(ModelStateMapper):
    """Select one tensor out of a stacked/batched input along dim 0
    (reference leaf SelectChild: e.g. pick expert i's slice from an
    (E, ...) tensor into its own key)."""

    def __init__(self, src: str, dst: str, index: int, dim: int = 0) -> None:
        self.src, self.dst, self.index, self.dim = src, dst, index, dim

    def state_dependency_groups(self) -> list[StateGroup]:
        return [StateGroup.of([self.src], [self.dst])]

    def apply_group(self, group, tensors):
        return {self.dst: tensors[self.src].select(self.dim, self.index).contiguous()}


class StackTensors(ModelStateMapper):
    def __init__(self, srcs: list[str], dst: str, dim: int = 0) -> None:
        self.srcs, self.dst, self.dim = list(srcs), dst, dim

    def state_dependency_groups(self):
        return [StateGroup.of(self.srcs, [self.dst])]

    def apply_group(self, group, tensors):
        return {self.dst: torch.stack([tensors[s] for s in self.srcs], dim=self.dim)}


class UnstackTensors(ModelStateMapper):
    def __init__(self, src: str, dsts: list[str], dim: int = 0) -> None:
        self.src, self.dsts, self.dim = src, list(dsts), dim

    def state_dependency_groups(self):
        return [StateGroup.of([self.src], self.dsts)]

    def apply_group(self, group, tensors):
        parts = tensors[self.src].unbind(self.dim)
        assert len(parts) == len(self.dsts)
        return {d: p.contiguous() for d, p in zip(self.dsts, parts)}


class ConcatenateTensors(ModelStateMapper):
    def __init__(self, srcs: list[str], dst: str, dim: int = 0) -> None:
        self.srcs, self.dst, self.dim = list(srcs), dst, dim

    def state_dependency_groups(self):
        return [StateGroup.of(self.srcs, [self.dst])]

    def apply_group(self, group, tensors):
        return {self.dst: torch.cat([tensors[s] for s in self.srcs], dim=self.dim)}


class ChunkTensors(ModelStateMapper):
    def __init__(self, src: str, dsts: list[str], dim: int = 0) -> None:
        self.src, self.dsts, self.dim = src, list(dsts), dim

    def state_dependency_groups(self):
        return [StateGroup.of([self.src], self.dsts)]

    def apply_group(self, group, tensors):
        parts = tensors[self.src].chunk(len(self.dsts), dim=self.dim)
        return {d: p.contiguous() for d, p in zip(self.dsts, parts)}


class SliceRows(ModelStateMapper):
    """Split one tensor into named variable-size row slices (e.g. the
    regular/special vocab split of SplitTokenEmbeddings)."""

    def __init__(self, src: str, dsts: list[tuple[str, int]], dim: int = 0) -> None:
        self.src, self.dsts, self.dim = src, list(dsts), dim

    def state_dependency_groups(self):
        return [StateGroup.of([self.src], [d for d, _ in self.dsts])]

    def apply_group(self, group, tensors):
        t = tensors[self.src]
        out = {}
        off = 0
        for name, size in self.dsts:
            out[name] = t.narrow(self.dim, off, size).contiguous()
            off += size
        return out


class Distribute(_SingleTensor):
    """Full local tensor -> DTensor, sliced locally (no communication;
    reference: leaf/dtensor.py Distribute with src_data_rank=None)."""

    def __init__(self, src: str, mesh: DeviceMesh, placements: tuple[Placement, ...],
                 dst: str | None = None):
        super().__init__(src, dst)
        self.mesh = mesh
        self.placements = placements

    def _fn(self, t):
        return distribute_tensor(t, self.mesh, self.placements, src_data_rank=None)


class GatherFullTensor(_SingleTensor):
    """DTensor -> full tensor (reference: leaf/dtensor.py:38)."""

    def _fn(self, t):
        if isinstance(t, DTensor):
            return t.full_tensor()
        return t


# ---- composition ------------------------------------------------------------


class Parallel(ModelStateMapper):
    """Disjoint union of mappers."""

    def __init__(self, *mappers: ModelStateMapper) -> None:
        self.mappers = list(mappers)
        self._group_owner: dict[StateGroup, ModelStateMapper] = {}
        for m in self.mappers:
            for g in m.state_dependency_groups():
                self._group_owner[g] = m

    def state_dependency_groups(self):
        return list(self._group_owner.keys())

    def apply_group(self, group, tensors):
        return self._group_owner[group].apply_group(group, tensors)


class Sequential(ModelStateMapper):
    """Chain mappers; keys not consumed downstream pass through (gap-filling
    auto-Identity) and chained groups merge into net input->output groups
    (reference: compose/sequential.py:12-70)."""

    def __init__(self, *mappers: ModelStateMapper) -> None:
        self.mappers = list(mappers)
        self._groups: list[StateGroup] = []
        self._plans: dict[StateGroup, list[tuple[ModelStateMapper, StateGroup]]] = {}
        self._build()

    def _build(self) -> None:
        # Greedy merge: start from the first mapper's groups (+ identity gaps
        # discovered later); push each through subsequent mappers.
        # chains: list of (inputs, outputs, plan) where plan = [(mapper, group)]
        chains: list[tuple[frozenset, frozenset, list]] = []
        first = self.mappers[0]
        for g in first.state_dependency_groups():
            chains.append((g.inputs, g.outputs, [(first, g)]))

        for m in self.mappers[1:]:
            new_chains: list[tuple[frozenset, frozenset, list]] = []
            groups = m.state_dependency_groups()
            consumed_chain: dict[int, bool] = {}
            for g in groups:
                # find chains providing any input of g
                feeding = [
                    i for i, (_, outs, _) in enumerate(chains) if outs & g.inputs
                ]
                in_keys: set = set(g.inputs)
                merged_inputs: set = set()
                plan: list = []
                provided: set = set()
                for i in feeding:
                    ins, outs, p = chains[i]
                    merged_inputs |= ins
                    plan += p
                    provided |= outs
                    consumed_chain[i] = True
                # inputs of g not provided by any chain come from the source
                merged_inputs |= in_keys - provided
                plan.append((m, g))
                # leftover chain outputs not consumed by g pass through
                leftover = provided - in_keys
                new_chains.append(
                    (frozenset(merged_inputs), frozenset(set(g.outputs) | leftover), plan)
                )
            # chains untouched by this mapper pass through unchanged
            for i, chain in enumerate(chains):
                if not consumed_chain.get(i):
                    new_chains.append(chain)
            chains = new_chains

        for ins, outs, plan in chains:
            group = StateGroup(ins, outs)
            self._groups.append(group)
            self._plans[group] = plan

    def state_dependency_groups(self):
        return list(self._groups)

    def apply_group(self, group, tensors):
        state = dict(tensors)
        for mapper, g in self._plans[group]:
            missing = [k for k in g.inputs if k not in state]
            if missing:
                raise KeyError(f"sequential group missing inputs {missing}")
            outs = mapper.apply_group(g, {k: state[k] for k in g.inputs})
            for k in g.inputs:
                if k not in group.outputs:
                    state.pop(k, None)
            state.update(outs)
        return {k: v for k, v in state.items() if k in group.outputs}


class PrefixScope(ModelStateMapper):
    """Apply `mapper` under a key prefix on both sides."""

    def __init__(self, prefix: str, mapper: ModelStateMapper) -> None:
        self.prefix = prefix
        self.mapper = mapper
        self._inner: dict[StateGroup, StateGroup] = {}
        for g in mapper.state_dependency_groups():
            outer = StateGroup(
                frozenset(prefix + k for k in g.inputs),
                frozenset(prefix + k for k in g.outputs),
            )
            self._inner[outer] = g

    def state_dependency_groups(self):
        return list(self._inner.keys())

    def apply_group(self, group, tensors):
        inner = self._inner[group]
        stripped = {k[len(self.prefix):]: v for k, v in tensors.items()}
        outs = self.mapper.apply_group(inner, stripped)
        return {self.prefix + k: v for k, v in outs.items()}


class Shard(ModelStateMapper):
    """Round-robin split of groups across ranks for parallel IO
    (reference: compose/shard.py:6-33)."""

    def __init__(self, mapper: ModelStateMapper, rank: int, world_size: int) -> None:
        self.mapper = mapper
        self.rank = rank
        self.world_size = world_size

    def state_dependency_groups(self):
        all_groups = self.mapper.state_dependency_groups()
        return [g for i, g in enumerate(all_groups) if i % self.world_size == self.rank]

    def apply_group(self, group, tensors):
        return self.mapper.apply_group(group, tensors)


def identity_mapper_from_module(module: nn.Module, prefix: str = "") -> ModelStateMapper:
    """One Identity group per state_dict entry (reference: adapters/module.py)."""
    keys = list(module.state_dict().keys())
    return Parallel(*[Identity(prefix + k) for k in keys])
