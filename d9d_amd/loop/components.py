"""Trainer components (reference: d9d/loop/component/)."""

import gc as _gc
import shutil
from pathlib import Path
from typing import Any

import torch
import torch.distributed as dist

from ..core.dist_context import DistributedContext
from ..core.offload import SleepTag, offload_tensor, onload_tensor
from ..internals.grad_norm import clip_grad_norm_distributed_
from ..internals.grad_sync import GradientSynchronizer


class BatchMaths:
    """global_batch = dp * microbatch * n_microbatches
    (reference: component/batch_maths.py:36-95)."""

    def __init__(self, global_batch_size: int, microbatch_size: int, dp: int, pp: int) -> None:
        self.global_batch_size = global_batch_size
        self.microbatch_size = microbatch_size
        self.dp = dp
        per_rank = global_batch_size // dp
        if per_rank * dp != global_batch_size:
            raise ValueError("global batch not divisible by dp degree")
        if per_rank % microbatch_size:
            raise ValueError("per-rank batch not divisible by microbatch size")
        self.num_microbatches = per_rank // microbatch_size
        # pipelining consumes all microbatches in one schedule step
        self.num_microbatches_pipelining = self.num_microbatches if pp > 1 else 1
        self.num_microbatches_gradient_accumulation = (
            1 if pp > 1 else self.num_microbatches
        )
        self.data_loader_batch_size = per_rank
        self.num_backward_calls = self.num_microbatches


class Stepper:
    """Step counter (reference: component/stepper.py)."""

    def __init__(self, total_steps: int) -> None:
        self.step = 0
        self.total_steps = total_steps

    def advance(self) -> None:
        self.step += 1

    @property
    def done(self) -> bool:
        return self.step >= self.total_steps

    def state_dict(self):
        return {"step": self.step}

    def load_state_dict(self, sd):
        self.step = sd["step"]


class GradientManager:
    """Owns the GradientSynchronizer + weighted loss scaling
    (reference: component/gradient_manager.py:46-211)."""

    def __init__(self, named_params, accumulation_steps: int, bucket_mb: int) -> None:
        self._named_params = list(named_params)
        self._accumulation_steps = accumulation_steps
        self._bucket_mb = bucket_mb
        self._sync: GradientSynchronizer | None = None
        self._loss_weight_total = 0.0

    def install(self) -> None:
        self._sync = GradientSynchronizer(
            self._named_params,
            accumulation_steps=self._accumulation_steps,
            bucket_bytes=self._bucket_mb * 1024 * 1024,
        )

    def add_loss_weight(self, weight: float) -> None:
        self._loss_weight_total += weight

    @property
    def loss_weight_total(self) -> float:
        return self._loss_weight_total

    def sync_and_scale(self, group=None, cp_group=None) -> None:
        """Wait grad comms; scale grads by 1/total weight (summed over dp
        and, under context parallelism, over cp — orthogonal groups, so two
        all-reduces compose to the product)."""
        if self._sync is not None:
            self._sync.wait()
        total = self._loss_weight_total
        for g in (group, cp_group):
            if dist.is_initialized() and g is not None:
                total = self._reduce_total(total, g)
        self._finish_scale(total)

    def _reduce_total(self, total, group):
        # The group is RCCL on GPU runs: the reduce tensor must live on
        # the same device as the gradients or the collective rejects it.
        device = None
        for _, p in self._named_params:
            if p.grad is not None:
                device = p.grad.device
                break
            if device is None:
                device = p.device
        t = torch.tensor([total], dtype=torch.float64, device=device)
        dist.all_reduce(t, group=group)
        return t.item()

    def _finish_scale(self, total: float) -> None:
        if total > 0:
            scale = 1.0 / total
            grads = []
            for _, p in self._named_params:
                if p.grad is not None:
                    g = p.grad
                    from torch.distributed.tensor import DTensor

                    grads.append(g.to_local() if isinstance(g, DTensor) else g)
            if grads:
                torch._foreach_mul_(grads, scale)
        self._loss_weight_total = 0.0

    def zero_grad(self) -> None:
        if self._sync is not None:
            self._sync.zero_grad()
        else:
            for _, p in self._named_params:
                p.grad = None

    def uninstall(self) -> None:
        if self._sync is not None:
            self._sync.remove()
            self._sync = None

    def offload(self, tags) -> None:
        if SleepTag.GRADS in tags:
            self.uninstall()

    def onload(self, tags) -> None:
        if SleepTag.GRADS in tags:
            self.install()


class GradientClipper:
    """Reference: component/gradient_clipper.py:41-86."""

    def __init__(self, params, max_norm: float, pp_group=None, enabled: bool = True):
        self.params = list(params)
        self.max_norm = max_norm
        self.pp_group = pp_group
        self.enabled = enabled
        self.last_norm: float | None = None

    def clip_and_log(self) -> float | None:
        if not self.enabled:
            return None
        norm = clip_grad_norm_distributed_(self.params, self.max_norm, self.pp_group)
        self.last_norm = float(norm)
        return self.last_norm


class Checkpointer:
    """DCP step checkpoints with rotation (reference: component/checkpointer.py:28-160)."""

    def __init__(self, save_dir: str | None, period_steps: int, num_to_keep: int):
        self.save_dir = Path(save_dir) if save_dir else None
        self.period_steps = period_steps
        self.num_to_keep = num_to_keep

    def _save_path(self, step: int) -> Path:
        return self.save_dir / f"save-{step}"

    def existing_checkpoints(self) -> list[int]:
        if self.save_dir is None or not self.save_dir.exists():
            return []
        steps = []
        for p in self.save_dir.glob("save-*"):
            try:
                steps.append(int(p.name.split("-")[1]))
            except (IndexError, ValueError):
                continue
        return sorted(steps)

    def checkpoint_if_needed(self, step: int, state: dict[str, Any]) -> bool:
        if self.save_dir is None or step == 0 or step % self.period_steps:
            return False
        self.save(step, state)
        return True

    def save(self, step: int, state: dict[str, Any]) -> None:
        import torch.distributed.checkpoint as dcp

        path = self._save_path(step)
        path.mkdir(parents=True, exist_ok=True)
        dcp.save(state, checkpoint_id=str(path))
        self._rotate()

    def load_last(self, state: dict[str, Any]) -> int | None:
        import torch.distributed.checkpoint as dcp

        steps = self.existing_checkpoints()
        if not steps:
            return None
        last = steps[-1]
        dcp.load(state, checkpoint_id=str(self._save_path(last)))
        return last

    def _rotate(self) -> None:
        steps = self.existing_checkpoints()
        is_main = not dist.is_initialized() or dist.get_rank() == 0
        while len(steps) > self.num_to_keep:
            victim = steps.pop(0)
            if is_main:
                shutil.rmtree(self._save_path(victim), ignore_errors=True)


class GarbageCollector:
    """Manual GC control (reference: component/garbage_collector.py:14-76)."""

    def __init__(self, period_steps: int) -> None:
        self.period_steps = period_steps
        self._was_enabled = _gc.isenabled()

    def install(self) -> None:
        _gc.disable()
        _gc.collect()

    def step(self, step: int) -> None:
        if self.period_steps and step % self.period_steps == 0:
            _gc.collect(1)

    def collect_full(self) -> None:
        _gc.collect()

    def uninstall(self) -> None:
        if self._was_enabled:
            _gc.enable()


class TimeoutManager:
    """Init vs step collective timeouts (reference: component/timeout_manager.py:15-66)."""

    def __init__(self, ctx: DistributedContext, init_s: float, step_s: float,
                 reapply_every: int = 50) -> None:
        self.ctx = ctx
        self.init_s = init_s
        self.step_s = step_s
        self.reapply_every = reapply_every

    def set_init(self) -> None:
        self.ctx.set_timeout(self.init_s)

    def set_periodic(self, step: int) -> None:
        if step % self.reapply_every == 0:
            self.ctx.set_timeout(self.step_s)


class TrainSleeper:
    """Sleep/wake orchestration (reference: component/train_sleeper.py:22-139)."""

    def __init__(self, ctx: DistributedContext, offloadables: list) -> None:
        self.ctx = ctx
        self.offloadables = offloadables
        self.sleeping: frozenset = frozenset()

    def sleep(self, tags: frozenset) -> None:
        if SleepTag.COMMS in tags:
            raise NotImplementedError("COMMS offload is not supported")
        self.ctx.wait_world()
        for obj in self.offloadables:
            obj.offload(tags)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
            torch.cuda.empty_cache()
        self.ctx.wait_world()
        self.sleeping = tags

    def wake(self) -> None:
        self.ctx.wait_world()
        for obj in reversed(self.offloadables):
            obj.onload(self.sleeping)
        self.ctx.wait_world()
        self.sleeping = frozenset()

    @property
    def is_sleeping(self) -> bool:
        return bool(self.sleeping)


class ModuleOffloader:
    """Offloadable wrapper over modules + optimizer state
    (reference: model_stage_factory.py TrackedModules + optimizer offload)."""

    def __init__(self, modules: list, optimizer=None) -> None:
        self.modules = modules
        self.optimizer = optimizer

    def offload(self, tags) -> None:
        if SleepTag.MODEL in tags:
            for m in self.modules:
                for t in list(m.parameters()) + list(m.buffers()):
                    offload_tensor(t)
        if SleepTag.OPTIMIZER in tags and self.optimizer is not None:
            for state in self.optimizer.state.values():
                for v in state.values():
                    if isinstance(v, torch.Tensor):
                        offload_tensor(v)

    def onload(self, tags) -> None:
        device = torch.device("cuda", torch.cuda.current_device()) if torch.cuda.is_available() else torch.device("cpu")
        if SleepTag.MODEL in tags:
            for m in self.modules:
                for t in list(m.parameters()) + list(m.buffers()):
                    onload_tensor(t, device)
        if SleepTag.OPTIMIZER in tags and self.optimizer is not None:
            for state in self.optimizer.state.values():
                for v in state.values():
                    if isinstance(v, torch.Tensor):
                        onload_tensor(v, device)
