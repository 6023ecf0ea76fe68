"""Example: Llama-3 dense training with the d9d_amd Trainer (+ optional LoRA).

The Llama-3 family (BASELINE.json's "Llama-3 70B TP=4 + SP + PP=2" config)
reuses the dense decoder stack without q/k norms. This example trains a small
config on synthetic data; swap `Llama3ModelParameters.tiny()` for
`.llama3_8b()` / `.llama3_70b()` and raise the mesh degrees on a real node.

Run single-GPU:     python example/llama3_finetune.py
Run multi-GPU (DP): torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
                        example/llama3_finetune.py
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
from d9d_amd.module.model.llama3 import Llama3ForCausalLM, Llama3ModelParameters
from d9d_amd.parallel import parallelize_replicate


class SyntheticLMDataset(Dataset):
    def __init__(self, vocab: int, seq_len: int, n: int = 1024, seed: int = 0):
        self.vocab, self.seq_len, self.n, self.seed = vocab, seq_len, n, seed

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        g = torch.Generator().manual_seed(self.seed + i)
        ids = torch.randint(0, self.vocab, (self.seq_len + 1,), generator=g)
        return {"input_ids": ids[:-1], "labels": ids[1:]}


class Llama3Provider(ModelProvider):
    def __init__(self, params: Llama3ModelParameters):
        self.params = params

    def initialize_model_stage(self, stage_info):
        return Llama3ForCausalLM(self.params, stage_info)

    def parallelize_model_stage(self, module, ctx):
        if ctx.world_size > 1:
            parallelize_replicate(module, ctx.mesh_for("dense"))
        return module

    def dump_hparams(self):
        return {"model": "llama3", "layers": self.params.num_hidden_layers}


class LMDatasetProvider(DatasetProvider):
    def __init__(self, params, seq_len=512):
        self.params, self.seq_len = params, seq_len

    def build_dataset(self, ctx):
        return SyntheticLMDataset(self.params.vocab_size, self.seq_len)


class FinetuneTask(TrainTask):
    def build_forward_inputs(self, batch):
        return {"input_ids": batch["input_ids"], "labels": batch["labels"]}

    def create_metrics(self):
        return {"loss_mean": WeightedMeanMetric()}

    def update_metrics(self, metrics, outputs, mb_inputs):
        metrics["loss_mean"].update(outputs["loss"].detach().mean(), 1.0)


def main():
    import os

    world = int(os.environ.get("WORLD_SIZE", "1"))
    params = Llama3ModelParameters.tiny()
    mesh = DeviceMeshParameters(data_parallel_replicate=world)
    config = TrainerConfig(
        batching=BatchingConfig(global_batch_size=4 * world, microbatch_size=2),
        logging=LoggingConfig(period_steps=5),
        total_steps=20,
    )
    trainer = TrainingConfigurator(
        config,
        mesh,
        Llama3Provider(params),
        LMDatasetProvider(params),
        AutoOptimizerProvider(OptimizerConfig(optimizer="adamw", lr=1e-4)),
        AutoLRSchedulerProvider(LRSchedulerConfig(warmup_steps=5, decay_steps=100)),
        FinetuneTask(),
    ).configure()
    trainer.train()


if __name__ == "__main__":
    main()
