"""Provider/task interfaces (reference: d9d/loop/control/).

Users implement these; the configurator wires them together.
"""

from abc import ABC, abstractmethod
from typing import Any

import torch
from torch import nn

from ..core.dist_context import DistributedContext
from ..metric.abc import Metric
from ..pipelining import PipelineStageInfo
from .event import EventBus


class ModelProvider(ABC):
    """Builds and parallelizes one pipeline-stage module
    (reference: control/model_provider.py:96-167)."""

    @abstractmethod
    def initialize_model_stage(self, stage_info: PipelineStageInfo) -> nn.Module:
        """Construct the stage module (called under torch.device('meta'))."""

    def parallelize_model_stage(
        self, module: nn.Module, ctx: DistributedContext
    ) -> nn.Module:
        return module

    def prepare_export_model_stage(self, module: nn.Module, stage_info):
        """Return a ModelStateMapper for exporting this stage (or None)."""
        return None

    def register_events(self, bus: EventBus) -> None:
        pass

    def dump_hparams(self) -> dict[str, Any]:
        return {}

    def source_checkpoint(self) -> str | None:
        """Optional path to stream initial weights from."""
        return None


class DatasetProvider(ABC):
    @abstractmethod
    def build_dataset(self, ctx: DistributedContext):
        """Return a torch Dataset of raw samples."""

    def collate(self, samples: list) -> Any:
        import torch.utils.data

        return torch.utils.data.default_collate(samples)


class OptimizerProvider(ABC):
    @abstractmethod
    def build_optimizer(self, named_params) -> torch.optim.Optimizer: ...


class LRSchedulerProvider(ABC):
    @abstractmethod
    def build_lr_scheduler(self, optimizer): ...


class BaseTask(ABC):
    def register_events(self, bus: EventBus) -> None:
        pass

    def create_metrics(self) -> dict[str, Metric]:
        return {}

    def dump_hparams(self) -> dict[str, Any]:
        return {}


class TrainTask(BaseTask):
    """Reference: control/task.py:75-299."""

    @abstractmethod
    def build_forward_inputs(self, batch: Any) -> dict[str, Any]:
        """Batch -> pipeline inputs dict (input_ids, labels, ...)."""

    def compute_loss(
        self, outputs: dict[str, torch.Tensor], mb_inputs: dict[str, Any]
    ) -> tuple[torch.Tensor, float]:
        """Last-stage microbatch outputs -> (loss, weight)."""
        return outputs["loss"].mean(), 1.0

    def update_metrics(self, metrics: dict[str, Metric], outputs, mb_inputs) -> None:
        pass


class InferenceTask(BaseTask):
    @abstractmethod
    def build_forward_inputs(self, batch: Any) -> dict[str, Any]: ...

    def process_outputs(self, outputs: dict[str, torch.Tensor], mb_inputs) -> Any:
        return outputs
