"""TrainingConfigurator + Trainer (reference: d9d/loop/run/train.py).

Wires mesh, data, pipeline schedule, optimizer, grad sync/clip, metrics,
checkpointing and events into the training loop:

    trainer = TrainingConfigurator(config, mesh_params, providers...).configure()
    trainer.train()
"""

import logging
from typing import Any

import torch
from torch.utils.data import DataLoader

from ..core.dist_context import DeviceMeshParameters, DistributedContext
from ..core.offload import SleepTag
from ..dataset import ShardedDataset
from ..internals.determinism import set_seeds
from ..internals.metric_collector import AsyncMetricCollector
from ..internals.profiling import Profiler
from ..pipelining.factory import build_schedule
from ..tracker import JsonlTracker, NullTracker
from . import event as ev
from .components import (
    BatchMaths,
    Checkpointer,
    GarbageCollector,
    GradientClipper,
    GradientManager,
    ModuleOffloader,
    Stepper,
    TimeoutManager,
    TrainSleeper,
)
from .config import TrainerConfig
from .control import (
    DatasetProvider,
    LRSchedulerProvider,
    ModelProvider,
    OptimizerProvider,
    TrainTask,
)
from .event import EventBus

logger = logging.getLogger("d9d_amd.train")


class _StatefulDict(dict):
    """Adapter: a dict of Statefuls usable as a DCP state root."""

    def state_dict(self):
        return {k: v.state_dict() for k, v in self.items()}

    def load_state_dict(self, sd):
        for k, v in self.items():
            if k in sd:
                v.load_state_dict(sd[k])


class _ModuleStates:
    def __init__(self, modules: dict[str, torch.nn.Module]):
        self.modules = modules

    def state_dict(self):
        return {k: m.state_dict() for k, m in self.modules.items()}

    def load_state_dict(self, sd):
        for k, m in self.modules.items():
            if k in sd:
                m.load_state_dict(sd[k])


class TrainingConfigurator:
    def __init__(
        self,
        config: TrainerConfig,
        mesh: DeviceMeshParameters,
        model_provider: ModelProvider,
        dataset_provider: DatasetProvider,
        optimizer_provider: OptimizerProvider,
        lr_scheduler_provider: LRSchedulerProvider,
        task: TrainTask,
    ) -> None:
        self.config = config
        self.mesh = mesh
        self.model_provider = model_provider
        self.dataset_provider = dataset_provider
        self.optimizer_provider = optimizer_provider
        self.lr_scheduler_provider = lr_scheduler_provider
        self.task = task

    def configure(self, device_type: str | None = None) -> "Trainer":
        cfg = self.config
        ctx = self.mesh.build(device_type=device_type)
        set_seeds(cfg.determinism.base_seed, ctx.pp_rank)

        bus = EventBus()
        self.task.register_events(bus)
        self.model_provider.register_events(bus)

        timeout = TimeoutManager(
            ctx, cfg.timeouts.init_timeout_seconds, cfg.timeouts.step_timeout_seconds
        )
        timeout.set_init()

        with bus.bounded(ev.TRAIN_CONFIGURE_PRE, ev.TRAIN_CONFIGURE_POST):
            trainer = self._build(ctx, bus, timeout)
        return trainer

    # -- build ----------------------------------------------------------------

    def _dp_degree(self, ctx: DistributedContext) -> int:
        d = self.mesh.domain_degrees()
        return d["dp"]

    def _build(self, ctx: DistributedContext, bus: EventBus, timeout) -> "Trainer":
        from ..ops.tunable import load_tuned_gemm_table

        load_tuned_gemm_table()  # pre-tuned hipBLASLt algo table for gfx950

        cfg = self.config
        dp = self._dp_degree(ctx)
        pp = self.mesh.pipeline_parallel
        maths = BatchMaths(
            cfg.batching.global_batch_size, cfg.batching.microbatch_size, dp, pp
        )

        # -- data -------------------------------------------------------------
        dataset = self.dataset_provider.build_dataset(ctx)
        if ctx.is_distributed and dp > 1:
            batch_mesh = ctx.mesh_for("batch")
            dp_rank = batch_mesh.get_local_rank("dp")
            dataset = ShardedDataset(dataset, dp_rank, dp)
        from .data import StatefulDataLoaderLite

        dp_rank = 0
        if ctx.is_distributed and dp > 1:
            dp_rank = ctx.mesh_for("batch").get_local_rank("dp")
        data_loader = StatefulDataLoaderLite(
            DataLoader(
                dataset,
                batch_size=maths.data_loader_batch_size,
                num_workers=cfg.data_loading.num_workers,
                pin_memory=cfg.data_loading.pin_memory,
                collate_fn=self.dataset_provider.collate,
                drop_last=True,
            ),
            dp_rank=dp_rank,
        )

        # -- model stages through the schedule factory ------------------------
        device = ctx.device
        pp_group = ctx.mesh_for("regular").get_group("pp") if ctx.is_distributed else None
        num_mb = (
            maths.num_microbatches if pp > 1 else maths.num_microbatches
        )

        def provider_fn(stage_info):
            # Reference flow (d9d/loop/component/model_stage_factory.py:218-249):
            # meta init -> parallelize on meta -> to_empty(device) ->
            # reset_parameters -> streamed load. A 70B-class stage never
            # materializes outside its own shard. Providers can opt out with
            # `meta_device_init = False` (e.g. when initialize copies real
            # pretrained weights directly).
            use_meta = getattr(self.model_provider, "meta_device_init", True)
            if use_meta:
                with torch.device("meta"):
                    module = self.model_provider.initialize_model_stage(stage_info)
                module = self.model_provider.parallelize_model_stage(module, ctx)
                module = module.to_empty(device=device)
            else:
                module = self.model_provider.initialize_model_stage(stage_info)
                module = self.model_provider.parallelize_model_stage(module, ctx)
                module = module.to(device=device)
            if hasattr(module, "reset_parameters"):
                module.reset_parameters()
            # parallelize steps that need real data (replicate's broadcast-
            # init) defer themselves when they ran on meta tensors
            for cb in getattr(module, "_d9d_post_materialize", []):
                cb(module)
            source = self.model_provider.source_checkpoint()
            if source:
                from ..model_state import load_model_state

                load_model_state(module, source)
            module.train()
            return module

        info = build_schedule(
            cfg.pipelining.schedule,
            provider_fn,
            num_microbatches=num_mb,
            device=device,
            pp_rank=ctx.pp_rank,
            pp_size=pp,
            pp_group=pp_group,
        )

        named_params: list = []
        modules_by_key: dict[str, torch.nn.Module] = {}
        for i, module in enumerate(info.modules):
            key = f"pp_{ctx.pp_rank}_stage_{i}"
            modules_by_key[key] = module
            for n, p in module.named_parameters():
                named_params.append((f"{key}.{n}", p))

        optimizer = self.optimizer_provider.build_optimizer(named_params)
        lr_scheduler = self.lr_scheduler_provider.build_lr_scheduler(optimizer)

        grad_manager = GradientManager(
            named_params, maths.num_backward_calls, cfg.gradient_sync.bucket_size_mb
        )
        dp_group = None
        if ctx.is_distributed and dp > 1:
            dp_group = ctx.mesh_for("batch").get_group("dp")
        cp_group = None
        cp_rank, cp_size = 0, self.mesh.context_parallel_shard
        if ctx.is_distributed and cp_size > 1:
            batch_mesh = ctx.mesh_for("batch")
            cp_group = batch_mesh.get_group("cp")
            cp_rank = batch_mesh.get_local_rank("cp")
        clipper = GradientClipper(
            [p for _, p in named_params],
            cfg.gradient_clipping.max_norm,
            pp_group=pp_group if pp > 1 else None,
            enabled=cfg.gradient_clipping.enabled,
        )

        metrics = self.task.create_metrics()
        collector = AsyncMetricCollector(metrics)
        tracker = (
            JsonlTracker(cfg.logging.tracker_dir)
            if cfg.logging.tracker == "jsonl" and ctx.is_main_process
            else NullTracker()
        )

        checkpointer = Checkpointer(
            cfg.checkpointing.save_dir,
            cfg.checkpointing.period_steps,
            cfg.checkpointing.num_to_keep,
        )
        profiler = Profiler(
            cfg.profiling.directory,
            rank_tag=f"rank{ctx.rank}",
            wait=cfg.profiling.wait,
            warmup=cfg.profiling.warmup,
            active=cfg.profiling.active,
            enabled=cfg.profiling.enabled,
        )
        gc = GarbageCollector(cfg.gc.period_steps)
        stepper = Stepper(cfg.total_steps)

        offloader = ModuleOffloader(info.modules, optimizer)
        sleeper = TrainSleeper(ctx, [grad_manager, offloader])

        return Trainer(
            config=cfg,
            ctx=ctx,
            bus=bus,
            task=self.task,
            schedule_info=info,
            data_loader=data_loader,
            maths=maths,
            optimizer=optimizer,
            lr_scheduler=lr_scheduler,
            grad_manager=grad_manager,
            clipper=clipper,
            collector=collector,
            tracker=tracker,
            checkpointer=checkpointer,
            profiler=profiler,
            gc=gc,
            stepper=stepper,
            timeout=timeout,
            sleeper=sleeper,
            modules_by_key=modules_by_key,
            dp_group=dp_group,
            cp_group=cp_group,
            cp_rank=cp_rank,
            cp_size=cp_size,
            model_provider=self.model_provider,
        )


class Trainer:
    def __init__(self, **kw: Any) -> None:
        self.config: TrainerConfig = kw["config"]
        self.ctx: DistributedContext = kw["ctx"]
        self.bus: EventBus = kw["bus"]
        self.task: TrainTask = kw["task"]
        self.schedule_info = kw["schedule_info"]
        self.data_loader = kw["data_loader"]
        self.maths: BatchMaths = kw["maths"]
        self.optimizer = kw["optimizer"]
        self.lr_scheduler = kw["lr_scheduler"]
        self.grad_manager: GradientManager = kw["grad_manager"]
        self.clipper: GradientClipper = kw["clipper"]
        self.collector: AsyncMetricCollector = kw["collector"]
        self.tracker = kw["tracker"]
        self.checkpointer: Checkpointer = kw["checkpointer"]
        self.profiler: Profiler = kw["profiler"]
        self.gc: GarbageCollector = kw["gc"]
        self.stepper: Stepper = kw["stepper"]
        self.timeout: TimeoutManager = kw["timeout"]
        self.sleeper: TrainSleeper = kw["sleeper"]
        self.modules_by_key = kw["modules_by_key"]
        self.dp_group = kw["dp_group"]
        self.cp_group = kw["cp_group"]
        self.cp_rank = kw["cp_rank"]
        self.cp_size = kw["cp_size"]
        self.model_provider: ModelProvider = kw["model_provider"]
        self._run = None
        self.last_losses: list[float] = []
        self.last_loss_weight: float = 0.0

    # -- checkpoint state schema (reference: loop/state.py:29-150) ------------

    def _job_state(self) -> _StatefulDict:
        # optimizer / lr-scheduler state is per-PIPELINE-STAGE (different
        # parameter shapes on each pp rank): the keys must be pp-qualified or
        # DCP's global planner sees one fqn with conflicting tensors
        # (AssertionError "item.index.fqn not in md").
        pp = self.ctx.pp_rank
        return _StatefulDict(
            **{
                "stepper": self.stepper,
                "tracked_modules": _ModuleStates(self.modules_by_key),
                "data_loader": self.data_loader,
                f"pp_{pp}_optimizer": self.optimizer,
                f"pp_{pp}_lr_scheduler": self.lr_scheduler,
                "metrics": self.collector,
                "tracker": self.tracker,
            }
        )

    # -- the loop --------------------------------------------------------------

    def _forward_backward(self, batch) -> None:
        inputs = self.task.build_forward_inputs(batch)
        inputs = {
            k: (v.to(self.ctx.device) if isinstance(v, torch.Tensor) else v)
            for k, v in inputs.items()
        }
        if self.cp_size > 1:
            # context parallelism: each cp rank trains on its contiguous
            # sequence chunk (dim 1 of every 2D+ tensor input); the model
            # offsets positions and ring attention spans the full context
            from ..parallel.context import shard_sequence

            inputs = {
                k: (
                    shard_sequence(v, self.cp_rank, self.cp_size, dim=1)
                    if isinstance(v, torch.Tensor) and v.ndim >= 2
                    else v
                )
                for k, v in inputs.items()
            }
        schedule = self.schedule_info.schedule
        schedule.configure_buffers(inputs)

        def loss_fn(mb, outputs, mb_inputs):
            loss, weight = self.task.compute_loss(outputs, mb_inputs)
            self.grad_manager.add_loss_weight(weight)
            self.task.update_metrics(self.collector.metrics, outputs, mb_inputs)
            return loss * weight

        with self.bus.bounded(ev.TRAIN_FORWARD_BACKWARD_PRE, ev.TRAIN_FORWARD_BACKWARD_POST):
            losses = schedule.step(inputs, loss_fn=loss_fn)
        self.last_losses = [float(l) for l in losses]
        # this rank's Σweight for the step — captured before sync_and_scale
        # resets it, so logging can report a per-unit-weight loss
        self.last_loss_weight = self.grad_manager.loss_weight_total

    def train(self) -> None:
        resumed = self.checkpointer.load_last(self._job_state())
        if resumed is not None:
            logger.info("resumed from step %d", resumed)

        self._run = self.tracker.new_run(
            self.config.run.name, self.config.run.description
        )
        hparams = {
            **self.config.run.hparams,
            **self.task.dump_hparams(),
            **self.model_provider.dump_hparams(),
        }
        if hparams:
            self._run.hparams(hparams)

        self.gc.install()
        self.grad_manager.install()
        self.profiler.open()

        data_iter = iter(self.data_loader)
        with self.bus.bounded(ev.TRAIN_RUN_PRE, ev.TRAIN_RUN_POST):
            while not self.stepper.done:
                try:
                    batch = next(data_iter)
                except StopIteration:
                    data_iter = iter(self.data_loader)
                    batch = next(data_iter)

                with self.bus.bounded(ev.TRAIN_STEP_PRE, ev.TRAIN_STEP_POST):
                    self._train_step(batch)

        self.profiler.close()
        self.grad_manager.uninstall()
        self.gc.uninstall()
        self._run.close()

    def _train_step(self, batch) -> None:
        self._forward_backward(batch)
        self.collector.trigger_sync()
        self.grad_manager.sync_and_scale(self.dp_group, self.cp_group)
        grad_norm = self.clipper.clip_and_log()
        with self.bus.bounded(ev.TRAIN_OPTIMIZER_STEP_PRE, ev.TRAIN_OPTIMIZER_STEP_POST):
            self.optimizer.step()
        self.lr_scheduler.step()
        self.stepper.advance()
        step = self.stepper.step

        if self._run is not None:
            self._run.set_step(step)
            if self.last_losses:
                # losses are weight-scaled (loss*weight per microbatch):
                # divide by the step's Σweight for a per-unit (per-token) loss
                denom = self.last_loss_weight if self.last_loss_weight > 0 else len(self.last_losses)
                self._run.scalar("loss", sum(self.last_losses) / denom)
            if grad_norm is not None:
                self._run.scalar("l2_grad_norm_total", grad_norm)
            self._run.scalar("lr", self.lr_scheduler.get_last_lr()[0])
            if step % self.config.logging.period_steps == 0:
                for name, value in self.collector.collect_results().items():
                    if isinstance(value, dict):
                        for k, v in value.items():
                            self._run.scalar(f"{name}/{k}", v)
                    else:
                        self._run.scalar(name, value)
                self.collector.reset()

        self.grad_manager.zero_grad()
        self.gc.step(step)
        self.profiler.step()
        self.timeout.set_periodic(step)
        with self.bus.bounded(ev.TRAIN_CHECKPOINT_PRE, ev.TRAIN_CHECKPOINT_POST):
            self.checkpointer.checkpoint_if_needed(step, self._job_state())

    # -- sleep/wake (reference: run/train.py:368-409) --------------------------

    def sleep(self, tags=frozenset({SleepTag.MODEL, SleepTag.OPTIMIZER, SleepTag.GRADS})) -> None:
        with self.bus.bounded(ev.TRAIN_SLEEP_PRE, ev.TRAIN_SLEEP_POST):
            self.sleeper.sleep(frozenset(tags))

    def wake(self) -> None:
        with self.bus.bounded(ev.TRAIN_WAKE_PRE, ev.TRAIN_WAKE_POST):
            self.sleeper.wake()

    @property
    def is_sleeping(self) -> bool:
        return self.sleeper.is_sleeping

    # -- export (reference: run/train.py:411-430) ------------------------------

    def export(self, path: str) -> None:
        from ..model_state import save_module_state, write_model_state_distributed

        for i, module in enumerate(self.schedule_info.modules):
            stage = self.schedule_info.stages[i]
            mapper = self.model_provider.prepare_export_model_stage(
                module, stage.stage_index
            )
            state = {
                k: v for k, v in module.state_dict().items()
            }
            if self.ctx.is_distributed:
                import torch.distributed as dist

                # one writer per pipeline stage: the first dp rank
                batch_mesh = self.ctx.mesh_for("batch")
                is_writer = all(
                    batch_mesh.get_local_rank(d) == 0
                    for d in ("dp", "cp", "tp")
                )
                if mapper is None:
                    from ..model_state.mapper import GatherFullTensor, Parallel

                    mapper = Parallel(*[GatherFullTensor(k) for k in state])
                write_model_state_distributed(
                    mapper, state, path, is_writer=is_writer
                )
            else:
                save_module_state(module, path, mapper)
