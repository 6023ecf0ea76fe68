"""GPU end-to-end: Trainer runs the MoE model with every HIP kernel engaged."""

import pytest
import torch

from d9d_amd.core.dist_context import DeviceMeshParameters
from d9d_amd.loop import TrainerConfig, TrainingConfigurator
from d9d_amd.loop.auto import (
    AutoLRSchedulerProvider,
    AutoOptimizerProvider,
    LRSchedulerConfig,
    OptimizerConfig,
)
from d9d_amd.loop.config import BatchingConfig
from tests.test_loop import _LMDatasetProvider, _LMTask


class _MoEProvider:
    def __init__(self, params):
        self.params = params

    def initialize_model_stage(self, stage_info):
        from d9d_amd.module.model.qwen3_moe import Qwen3MoEForCausalLM

        return Qwen3MoEForCausalLM(self.params, stage_info, dtype=torch.bfloat16)

    def parallelize_model_stage(self, module, ctx):
        return module

    def register_events(self, bus):
        pass

    def dump_hparams(self):
        return {}

    def source_checkpoint(self):
        return None

    def prepare_export_model_stage(self, module, stage_info):
        return None


@pytest.mark.gpu
def test_trainer_gpu_moe_two_steps(tmp_path):
    from d9d_amd.module.model.qwen3_moe import Qwen3MoEModelParameters

    params = Qwen3MoEModelParameters(
        hidden_size=128,
        intermediate_size=96,
        num_experts=16,
        experts_top_k=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        head_dim=32,
        num_hidden_layers=2,
        split_vocab_size={"regular": 1000, "special": 24},
    )
    config = TrainerConfig(
        batching=BatchingConfig(global_batch_size=8, microbatch_size=4),
        total_steps=2,
    )
    trainer = TrainingConfigurator(
        config,
        DeviceMeshParameters(),
        _MoEProvider(params),
        _LMDatasetProvider(params),
        AutoOptimizerProvider(OptimizerConfig(optimizer="stochastic_adamw", lr=1e-3)),
        AutoLRSchedulerProvider(LRSchedulerConfig(decay_steps=10)),
        _LMTask(),
    ).configure(device_type="cuda")
    trainer.train()
    assert trainer.stepper.step == 2
    assert trainer.last_losses and all(
        torch.isfinite(torch.tensor(trainer.last_losses))
    )
    trainer.export(str(tmp_path / "export"))
    from d9d_amd.model_state import read_model_state

    keys = dict(read_model_state(tmp_path / "export"))
    assert any("experts" in k for k in keys)
