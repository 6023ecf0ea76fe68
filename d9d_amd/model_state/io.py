"""Streaming safetensors IO over the mapper DAG (reference: d9d/model_state/io/).

Format: `model-XXXXX-of-YYYYY.safetensors` shards + `model.safetensors.index.json`
({"metadata": {...}, "weight_map": {key: filename}}) — HF-compatible.

Reader: per-file load plan, fires a mapper group as soon as all inputs are in
memory, evicts consumed inputs. Writer: buffers group outputs into <=
`shard_size_gb` shards; the distributed variant writes per-rank temp shards
and merges indices on rank 0 (reference: io/writer.py:61-309).
"""

import json
from pathlib import Path
from typing import Iterator

import torch
import torch.distributed as dist
from safetensors import safe_open
from safetensors.torch import save_file
from torch import nn
from torch.distributed.tensor import DTensor

from .mapper import Distribute, ModelStateMapper, Parallel, StateGroup

INDEX_NAME = "model.safetensors.index.json"


def _read_index(path: Path) -> dict[str, str]:
    index_file = path / INDEX_NAME
    if index_file.exists():
        with open(index_file) as f:
            return json.load(f)["weight_map"]
    # single-file checkpoints
    single = path / "model.safetensors"
    if single.exists():
        with safe_open(str(single), framework="pt") as f:
            return {k: "model.safetensors" for k in f.keys()}
    raise FileNotFoundError(f"no {INDEX_NAME} or model.safetensors under {path}")


def read_model_state(path: str | Path) -> Iterator[tuple[str, torch.Tensor]]:
    """Stream (key, tensor) pairs file by file."""
    path = Path(path)
    weight_map = _read_index(path)
    by_file: dict[str, list[str]] = {}
    for key, fname in weight_map.items():
        by_file.setdefault(fname, []).append(key)
    for fname in sorted(by_file):
        with safe_open(str(path / fname), framework="pt") as f:
            for key in by_file[fname]:
                yield key, f.get_tensor(key)


class _StreamingApplier:
    """Fires mapper groups as their inputs arrive; evicts consumed inputs."""

    def __init__(self, mapper: ModelStateMapper):
        self.mapper = mapper
        self.groups = mapper.state_dependency_groups()
        self.pending: dict[str, torch.Tensor] = {}
        self.waiting: dict[str, list[StateGroup]] = {}
        self.remaining: dict[StateGroup, set] = {}
        for g in self.groups:
            self.remaining[g] = set(g.inputs)
            for k in g.inputs:
                self.waiting.setdefault(k, []).append(g)

    def offer(self, key: str, tensor: torch.Tensor) -> Iterator[dict]:
        if key not in self.waiting:
            return
        self.pending[key] = tensor
        for g in self.waiting[key]:
            rem = self.remaining.get(g)
            if rem is None:
                continue
            rem.discard(key)
            if not rem:
                del self.remaining[g]
                outs = self.mapper.apply_group(
                    g, {k: self.pending[k] for k in g.inputs}
                )
                yield outs
        # evict inputs no UNFIRED group needs (arrived inputs of pending
        # groups must stay resident until the group fires)
        still_needed = {k for g in self.remaining for k in g.inputs}
        for k in list(self.pending):
            if k not in still_needed:
                del self.pending[k]

    def unfired_groups(self) -> list[StateGroup]:
        return list(self.remaining.keys())


def write_model_state(
    mapper: ModelStateMapper,
    source: Iterator[tuple[str, torch.Tensor]] | dict[str, torch.Tensor],
    path: str | Path,
    shard_size_gb: float = 4.0,
    file_prefix: str = "model",
    write_index: bool = True,
    persist: bool = True,
) -> dict[str, str]:
    """Run `source` through `mapper`, writing outputs as safetensors shards.

    `persist=False` runs the full traversal WITHOUT buffering or writing:
    mapper leaves may issue collectives (DTensor `full_tensor`), so in a
    distributed export every rank must traverse every group in the same
    order even when only some ranks write."""
    path = Path(path)
    if persist:
        path.mkdir(parents=True, exist_ok=True)
    applier = _StreamingApplier(mapper)
    items = source.items() if isinstance(source, dict) else source

    cap_bytes = int(shard_size_gb * (1 << 30))
    buffer: dict[str, torch.Tensor] = {}
    buffered = 0
    shards: list[dict[str, torch.Tensor]] = []

    def flush():
        nonlocal buffer, buffered
        if buffer:
            shards.append(buffer)
            buffer = {}
            buffered = 0

    for key, tensor in items:
        for outs in applier.offer(key, tensor):
            for ok, ov in outs.items():
                if isinstance(ov, DTensor):
                    ov = ov.full_tensor()  # collective: runs on EVERY rank
                if not persist:
                    continue
                ov = ov.detach().cpu().contiguous()
                buffer[ok] = ov
                buffered += ov.numel() * ov.element_size()
                if buffered >= cap_bytes:
                    flush()
    flush()
    if applier.unfired_groups():
        missing = sorted(
            k for g in applier.unfired_groups() for k in g.inputs
        )[:10]
        raise ValueError(f"model_state write: missing inputs for groups, e.g. {missing}")

    if not persist:
        return {}
    total = len(shards)
    weight_map: dict[str, str] = {}
    for i, shard in enumerate(shards):
        fname = f"{file_prefix}-{i + 1:05d}-of-{total:05d}.safetensors"
        save_file(shard, str(path / fname))
        for k in shard:
            weight_map[k] = fname
    if write_index:
        with open(path / INDEX_NAME, "w") as f:
            json.dump({"metadata": {}, "weight_map": weight_map}, f, indent=2)
    return weight_map


def write_model_state_distributed(
    mapper: ModelStateMapper,
    source,
    path: str | Path,
    shard_size_gb: float = 4.0,
    group=None,
    is_writer: bool = True,
) -> None:
    """Every writer rank streams its (Shard-ed) groups to rank-tagged shards;
    rank 0 merges the index (reference: io/writer.py:252-309)."""
    path = Path(path)
    rank = dist.get_rank(group) if dist.is_initialized() else 0
    # EVERY rank traverses the mapper (DTensor gathers inside are
    # collectives); only writer ranks persist shards.
    maps: dict[str, str] = write_model_state(
        mapper, source, path,
        shard_size_gb=shard_size_gb,
        file_prefix=f"model-rank{rank}",
        write_index=False,
        persist=is_writer,
    )
    if dist.is_initialized():
        from ..core.dist_ops import all_gather_object

        all_maps = all_gather_object(maps, group=group)
    else:
        all_maps = [maps]
    if rank == 0:
        merged: dict[str, str] = {}
        for m in all_maps:
            merged.update(m)
        with open(path / INDEX_NAME, "w") as f:
            json.dump({"metadata": {}, "weight_map": merged}, f, indent=2)
    if dist.is_initialized():
        dist.barrier()


def load_model_state(
    module: nn.Module,
    path: str | Path,
    mapper: ModelStateMapper | None = None,
    strict: bool = False,
) -> list[str]:
    """Stream a checkpoint into `module` through `mapper`.

    DTensor parameters receive the full tensor via local slicing (the
    auto-Distribute injection of reference io/module_reader.py:27-84).
    Returns the list of module keys that were loaded.
    """
    state = module.state_dict()
    if mapper is None:
        from .mapper import Identity

        mapper = Parallel(*[Identity(k) for k in state])

    loaded: list[str] = []
    applier = _StreamingApplier(mapper)
    with torch.no_grad():
        for key, tensor in read_model_state(path):
            for outs in applier.offer(key, tensor):
                for ok, ov in outs.items():
                    if ok not in state:
                        if strict:
                            raise KeyError(f"unexpected checkpoint key {ok}")
                        continue
                    target = state[ok]
                    if isinstance(target, DTensor) and not isinstance(ov, DTensor):
                        from torch.distributed.tensor import distribute_tensor

                        ov = distribute_tensor(
                            ov.to(target.to_local().dtype),
                            target.device_mesh,
                            target.placements,
                            src_data_rank=None,
                        )
                        target._local_tensor.copy_(ov._local_tensor)
                    else:
                        target.copy_(ov.to(target.dtype))
                    loaded.append(ok)
    if strict:
        missing = sorted(set(state) - set(loaded))
        if missing:
            raise KeyError(f"missing checkpoint keys: {missing[:10]}...")
    return loaded


def save_module_state(
    module: nn.Module,
    path: str | Path,
    mapper: ModelStateMapper | None = None,
    shard_size_gb: float = 4.0,
) -> None:
    """Convenience: module state_dict -> (mapper) -> safetensors shards;
    DTensors are gathered to full tensors first."""
    state = {}
    for k, v in module.state_dict().items():
        state[k] = v.full_tensor() if isinstance(v, DTensor) else v
    if mapper is None:
        from .mapper import Identity

        mapper = Parallel(*[Identity(k) for k in state])
    write_model_state(mapper, state, path, shard_size_gb=shard_size_gb)
