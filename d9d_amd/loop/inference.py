"""InferenceConfigurator — forward-only loop (reference: d9d/loop/run/inference.py:55-260)."""

from typing import Any, Callable

import torch
from torch.utils.data import DataLoader

from ..core.dist_context import DeviceMeshParameters
from ..dataset import ShardedDataset
from ..internals.determinism import set_seeds
from ..pipelining.factory import PipelineScheduleInferenceConfig, build_schedule
from . import event as ev
from .config import TrainerConfig
from .control import DatasetProvider, InferenceTask, ModelProvider
from .event import EventBus


class InferenceRunner:
    def __init__(self, ctx, bus, schedule_info, data_loader, task, device) -> None:
        self.ctx = ctx
        self.bus = bus
        self.schedule_info = schedule_info
        self.data_loader = data_loader
        self.task = task
        self.device = device

    @torch.no_grad()
    def run(self, on_result: Callable[[Any], None] | None = None) -> list:
        results: list = []
        with self.bus.bounded(ev.INFER_RUN_PRE, ev.INFER_RUN_POST):
            for batch in self.data_loader:
                with self.bus.bounded(ev.INFER_BATCH_PRE, ev.INFER_BATCH_POST):
                    inputs = self.task.build_forward_inputs(batch)
                    inputs = {
                        k: (v.to(self.device) if isinstance(v, torch.Tensor) else v)
                        for k, v in inputs.items()
                    }
                    schedule = self.schedule_info.schedule
                    schedule.configure_buffers(inputs)

                    def loss_fn(mb, outputs, mb_inputs):
                        processed = self.task.process_outputs(outputs, mb_inputs)
                        results.append(processed)
                        if on_result is not None:
                            on_result(processed)
                        return None

                    schedule.step(inputs, loss_fn=loss_fn)
        return results


class InferenceConfigurator:
    def __init__(
        self,
        config: TrainerConfig,
        mesh: DeviceMeshParameters,
        model_provider: ModelProvider,
        dataset_provider: DatasetProvider,
        task: InferenceTask,
    ) -> None:
        self.config = config
        self.mesh = mesh
        self.model_provider = model_provider
        self.dataset_provider = dataset_provider
        self.task = task

    def configure(self, device_type: str | None = None) -> InferenceRunner:
        cfg = self.config
        ctx = self.mesh.build(device_type=device_type)
        set_seeds(cfg.determinism.base_seed, ctx.pp_rank)
        bus = EventBus()
        self.task.register_events(bus)
        self.model_provider.register_events(bus)

        dp = self.mesh.domain_degrees()["dp"]
        pp = self.mesh.pipeline_parallel
        dataset = self.dataset_provider.build_dataset(ctx)
        if ctx.is_distributed and dp > 1:
            dp_rank = ctx.mesh_for("batch").get_local_rank("dp")
            dataset = ShardedDataset(dataset, dp_rank, dp)
        per_rank = cfg.batching.global_batch_size // dp
        num_mb = max(per_rank // cfg.batching.microbatch_size, 1)
        data_loader = DataLoader(
            dataset,
            batch_size=per_rank,
            collate_fn=self.dataset_provider.collate,
            drop_last=False,
        )

        device = ctx.device
        pp_group = ctx.mesh_for("regular").get_group("pp") if ctx.is_distributed else None

        def provider_fn(stage_info):
            module = self.model_provider.initialize_model_stage(stage_info)
            module = self.model_provider.parallelize_model_stage(module, ctx)
            module = module.to(device=device)
            if hasattr(module, "reset_parameters"):
                module.reset_parameters()
            source = self.model_provider.source_checkpoint()
            if source:
                from ..model_state import load_model_state

                load_model_state(module, source)
            module.eval()
            return module

        with bus.bounded(ev.INFER_CONFIGURE_PRE, ev.INFER_CONFIGURE_POST):
            info = build_schedule(
                PipelineScheduleInferenceConfig(),
                provider_fn,
                num_microbatches=num_mb,
                device=device,
                pp_rank=ctx.pp_rank,
                pp_size=pp,
                pp_group=pp_group,
            )
        return InferenceRunner(ctx, bus, info, data_loader, self.task, device)
