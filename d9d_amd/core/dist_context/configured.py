"""DistributedContext — owns process groups, device binding and named meshes.

MI355X-native analog of the reference's DistributedContext
(d9d/core/dist_context/configured.py:34-171): one process per GPU, RCCL over
xGMI via torch.distributed's "nccl" backend (NCCL==RCCL on ROCm), five named
DeviceMesh domains over the same world, rank-qualified logging, timeout
control and world barriers.
"""

import datetime
import logging
import os
from contextlib import contextmanager
from typing import Iterator

import torch
import torch.distributed as dist
from torch.distributed.device_mesh import DeviceMesh, init_device_mesh

from .domains import ALL_DOMAINS
from .params import DeviceMeshParameters

logger = logging.getLogger("d9d_amd")


def _resolve_device_type() -> str:
    return "cuda" if torch.cuda.is_available() else "cpu"


def _backend_for(device_type: str) -> str:
    # "nccl" IS RCCL on ROCm builds of PyTorch.
    return "nccl" if device_type == "cuda" else "gloo"


class DistributedContext:
    """Topology handle: meshes per domain + rank/device utilities."""

    def __init__(
        self,
        params: DeviceMeshParameters,
        device_type: str,
        meshes: dict[str, DeviceMesh],
        rank: int,
        local_rank: int,
        world_size: int,
    ) -> None:
        self.params = params
        self.device_type = device_type
        self._meshes = meshes
        self.rank = rank
        self.local_rank = local_rank
        self.world_size = world_size

    # -- construction ---------------------------------------------------------

    @classmethod
    def create(
        cls,
        params: DeviceMeshParameters,
        device_type: str | None = None,
    ) -> "DistributedContext":
        device_type = device_type or _resolve_device_type()

        env_world = int(os.environ.get("WORLD_SIZE", "1"))
        distributed = env_world > 1 or dist.is_initialized() or params.world_size > 1

        if not distributed:
            # Local mode: no process groups; mesh-free code paths only.
            if device_type == "cuda":
                torch.cuda.set_device(0)
            ctx = cls(params, device_type, {}, rank=0, local_rank=0, world_size=1)
            ctx._install_log_prefix()
            return ctx

        if not dist.is_initialized():
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29511")
            dist.init_process_group(
                backend=_backend_for(device_type),
                timeout=datetime.timedelta(seconds=params.init_timeout_seconds),
            )

        rank = dist.get_rank()
        world_size = dist.get_world_size()
        local_rank = int(os.environ.get("LOCAL_RANK", rank % max(1, torch.cuda.device_count() or 1)))

        if world_size != params.world_size:
            raise ValueError(
                f"mesh degrees multiply to {params.world_size} but world size is {world_size}"
            )

        if device_type == "cuda":
            torch.cuda.set_device(local_rank)

        shapes = params.domain_shapes()
        meshes: dict[str, DeviceMesh] = {}
        for dom in ALL_DOMAINS:
            dim_names, shape = shapes[dom.name]
            meshes[dom.name] = init_device_mesh(
                device_type, shape, mesh_dim_names=dim_names
            )

        ctx = cls(params, device_type, meshes, rank, local_rank, world_size)
        ctx._install_log_prefix()
        return ctx

    def _install_log_prefix(self) -> None:
        if not logger.handlers:
            handler = logging.StreamHandler()
            handler.setFormatter(
                logging.Formatter(
                    f"[rank {self.rank}/{self.world_size}] %(asctime)s %(levelname)s %(message)s"
                )
            )
            logger.addHandler(handler)
            logger.setLevel(logging.INFO)

    # -- meshes ---------------------------------------------------------------

    @property
    def is_distributed(self) -> bool:
        return bool(self._meshes)

    def mesh_for(self, domain: str) -> DeviceMesh:
        if domain not in self._meshes:
            raise KeyError(
                f"unknown or unavailable mesh domain {domain!r} "
                f"(available: {sorted(self._meshes)}; local mode has none)"
            )
        return self._meshes[domain]

    @property
    def device(self) -> torch.device:
        if self.device_type == "cuda":
            return torch.device("cuda", self.local_rank)
        return torch.device("cpu")

    @property
    def is_main_process(self) -> bool:
        return self.rank == 0

    @property
    def is_local_main_process(self) -> bool:
        return self.local_rank == 0

    # -- pipeline coordinates -------------------------------------------------

    @property
    def pp_rank(self) -> int:
        if not self.is_distributed:
            return 0
        return int(self.mesh_for("regular").get_local_rank("pp"))

    @property
    def pp_size(self) -> int:
        return self.params.pipeline_parallel

    # -- synchronization ------------------------------------------------------

    def wait_world(self) -> None:
        if not self.is_distributed:
            return
        if self.device_type == "cuda":
            dist.barrier(device_ids=[self.local_rank])
        else:
            dist.barrier()

    def set_timeout(self, seconds: float) -> None:
        """Re-arm the collective timeout on every process group."""
        if not self.is_distributed:
            return
        timeout = datetime.timedelta(seconds=seconds)
        seen: set[int] = set()
        groups = [dist.group.WORLD]
        for mesh in self._meshes.values():
            for dim in range(mesh.ndim):
                groups.append(mesh.get_group(dim))
        for group in groups:
            if group is None or id(group) in seen:
                continue
            seen.add(id(group))
            try:
                dist.distributed_c10d._set_pg_timeout(timeout, group)
            except (RuntimeError, AttributeError, ValueError):
                # Gloo on some builds does not support re-arming; non-fatal.
                pass

    @contextmanager
    def main_process_first(self) -> Iterator[None]:
        """Main rank runs the body first, others after a barrier (e.g. cache fill)."""
        if self.is_distributed and not self.is_main_process:
            self.wait_world()
        try:
            yield
        finally:
            if self.is_distributed and self.is_main_process:
                self.wait_world()

    def destroy(self) -> None:
        if dist.is_initialized():
            dist.destroy_process_group()
