"""Example: Qwen3-MoE pretraining with the d9d_amd Trainer.

Mirrors the reference entry script (reference: example/qwen3_moe/pretrain.py):
providers + a pydantic-validated config -> TrainingConfigurator -> train().
Run single-GPU:     python example/qwen3_moe_pretrain.py
Run multi-GPU (DP): torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
                        example/qwen3_moe_pretrain.py
"""

import pathlib
import sys

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))

import torch
from torch.utils.data import Dataset

from d9d_amd.core.dist_context import DeviceMeshParameters
from d9d_amd.loop import TrainerConfig, TrainingConfigurator
from d9d_amd.loop.auto import (
    AutoLRSchedulerProvider,
    AutoOptimizerProvider,
    LRSchedulerConfig,
    OptimizerConfig,
)
from d9d_amd.loop.config import BatchingConfig, LoggingConfig
from d9d_amd.loop.control import DatasetProvider, ModelProvider, TrainTask
from d9d_amd.metric import WeightedMeanMetric
from d9d_amd.module.model.qwen3_moe import (
    Qwen3MoEForCausalLM,
    Qwen3MoEModelParameters,
)
from d9d_amd.parallel import parallelize_expert_parallel, parallelize_replicate


class SyntheticLMDataset(Dataset):
    """Random-token stand-in for a tokenized pretraining corpus."""

    def __init__(self, vocab: int, seq_len: int, n: int = 4096, seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        self.data = torch.randint(0, vocab, (n, seq_len + 1), generator=g)

    def __len__(self):
        return len(self.data)

    def __getitem__(self, i):
        return self.data[i]


class Qwen3MoEProvider(ModelProvider):
    def __init__(self, params: Qwen3MoEModelParameters):
        self.params = params

    def initialize_model_stage(self, stage_info):
        # StochasticAdamW requires bf16 parameters (fp32 moments + SR writes)
        return Qwen3MoEForCausalLM(self.params, stage_info).to(torch.bfloat16)

    def parallelize_model_stage(self, module, ctx):
        if not ctx.is_distributed:
            return module
        if ctx.params.expert_parallel > 1:
            parallelize_expert_parallel(module, ctx.mesh_for("expert"))
        parallelize_replicate(module, ctx.mesh_for("dense"))
        return module

    def dump_hparams(self):
        return {"model": "qwen3-moe", "layers": self.params.num_hidden_layers}


class LMDatasetProvider(DatasetProvider):
    def __init__(self, params, seq_len=2048):
        self.params = params
        self.seq_len = seq_len

    def build_dataset(self, ctx):
        return SyntheticLMDataset(self.params.vocab_size, self.seq_len)


class PretrainTask(TrainTask):
    def build_forward_inputs(self, batch):
        return {"input_ids": batch[:, :-1], "labels": batch[:, 1:]}

    def create_metrics(self):
        return {"train_loss": WeightedMeanMetric()}

    def update_metrics(self, metrics, outputs, mb_inputs):
        metrics["train_loss"].update(outputs["loss"].detach().mean(), 1.0)


def main():
    import os

    world = int(os.environ.get("WORLD_SIZE", "1"))
    # full pretrain config on GPUs; tiny config for CPU smoke runs
    params = (
        Qwen3MoEModelParameters.example_pretrain()
        if torch.cuda.is_available()
        else Qwen3MoEModelParameters.tiny()
    )
    mesh = DeviceMeshParameters(
        data_parallel_replicate=world,
        expert_parallel=world if world > 1 else 1,
    )
    config = TrainerConfig(
        batching=BatchingConfig(global_batch_size=16 * world, microbatch_size=8),
        logging=LoggingConfig(period_steps=5, tracker="jsonl", tracker_dir="./logs"),
        total_steps=20,
    )
    trainer = TrainingConfigurator(
        config,
        mesh,
        Qwen3MoEProvider(params),
        # short sequences for the CPU demo path; 2048+ on GPUs
        LMDatasetProvider(params, seq_len=2048 if torch.cuda.is_available() else 256),
        AutoOptimizerProvider(OptimizerConfig(optimizer="stochastic_adamw", lr=3e-4)),
        AutoLRSchedulerProvider(LRSchedulerConfig(warmup_steps=5, decay_steps=100)),
        PretrainTask(),
    ).configure()
    trainer.train()
    trainer.export("./export")


if __name__ == "__main__":
    main()
