"""Pydantic config tree (reference: d9d/loop/config/config.py:11-225)."""

from typing import Literal, Optional

from pydantic import BaseModel, Field

from ..pipelining.factory import PipelineScheduleConfig, PipelineScheduleGPipeConfig


class MeshConfig(BaseModel):
    pipeline_parallel: int = 1
    data_parallel_replicate: int = 1
    data_parallel_shard: int = 1
    context_parallel_shard: int = 1
    context_parallel_replicate: int = 1
    tensor_parallel: int = 1
    expert_parallel: int = 1


class RunConfig(BaseModel):
    name: str = "run"
    description: str = ""
    hparams: dict = Field(default_factory=dict)


class BatchingConfig(BaseModel):
    global_batch_size: int
    microbatch_size: int


class DataLoadingConfig(BaseModel):
    num_workers: int = 0
    pin_memory: bool = False
    persistent_workers: bool = False


class StepActionPeriod(BaseModel):
    period_steps: int = 1
    offset_steps: int = 0

    def should_act(self, step: int) -> bool:
        return step >= self.offset_steps and (step - self.offset_steps) % self.period_steps == 0


class LoggingConfig(BaseModel):
    period_steps: int = 10
    tracker: Literal["null", "jsonl"] = "null"
    tracker_dir: str = "./logs"


class PipeliningConfig(BaseModel):
    schedule: PipelineScheduleConfig = Field(
        default_factory=PipelineScheduleGPipeConfig
    )


class CheckpointingConfig(BaseModel):
    save_dir: Optional[str] = None
    period_steps: int = 1000
    num_to_keep: int = 3


class GradientClippingConfig(BaseModel):
    max_norm: float = 1.0
    enabled: bool = True


class GradientSyncConfig(BaseModel):
    bucket_size_mb: int = 64


class DeterminismConfig(BaseModel):
    base_seed: int = 1337


class GcConfig(BaseModel):
    period_steps: int = 10


class ProfilingConfig(BaseModel):
    enabled: bool = False
    directory: str = "./profiles"
    wait: int = 1
    warmup: int = 2
    active: int = 3


class TimeoutConfig(BaseModel):
    init_timeout_seconds: float = 600.0
    step_timeout_seconds: float = 120.0


class TrainerConfig(BaseModel):
    run: RunConfig = Field(default_factory=RunConfig)
    batching: BatchingConfig
    data_loading: DataLoadingConfig = Field(default_factory=DataLoadingConfig)
    logging: LoggingConfig = Field(default_factory=LoggingConfig)
    pipelining: PipeliningConfig = Field(default_factory=PipeliningConfig)
    checkpointing: CheckpointingConfig = Field(default_factory=CheckpointingConfig)
    gradient_clipping: GradientClippingConfig = Field(default_factory=GradientClippingConfig)
    gradient_sync: GradientSyncConfig = Field(default_factory=GradientSyncConfig)
    determinism: DeterminismConfig = Field(default_factory=DeterminismConfig)
    gc: GcConfig = Field(default_factory=GcConfig)
    profiling: ProfilingConfig = Field(default_factory=ProfilingConfig)
    timeouts: TimeoutConfig = Field(default_factory=TimeoutConfig)
    total_steps: int = 1000
