"""PEFT inject/merge round-trip tests (reference: test/d9d_test/peft)."""

import torch
from torch import nn

from d9d_amd.peft import (
    FullTuneMethod,
    LoRAGroupedLinear,
    LoRALinear,
    LoRAMethod,
    PeftStack,
    inject_peft_and_freeze,
)


def test_lora_inject_freeze_and_merge_linear():
    torch.manual_seed(0)
    model = nn.Sequential()
    model.add_module("q_proj", nn.Linear(8, 8, bias=False))
    model.add_module("act", nn.ReLU())

    model, mapper = inject_peft_and_freeze(model, LoRAMethod(rank=2, alpha=4))
    assert isinstance(model.q_proj, LoRALinear)
    trainable = [n for n, p in model.named_parameters() if p.requires_grad]
    assert all("lora" in n for n in trainable) and trainable

    x = torch.randn(3, 8)
    with torch.no_grad():
        model.q_proj.lora_B.normal_()  # make the adapter non-trivial
        y_adapter = model(x)
    merged = LoRAMethod(rank=2, alpha=4).merge(model)
    assert isinstance(merged.q_proj, nn.Linear)
    with torch.no_grad():
        y_merged = merged(x)
    torch.testing.assert_close(y_adapter, y_merged, rtol=1e-5, atol=1e-6)


def test_lora_grouped_linear():
    from d9d_amd.module.block.moe.grouped_linear import GroupedLinear

    torch.manual_seed(1)
    gl = GroupedLinear(num_experts=4, in_features=8, out_features=6)
    gl.reset_parameters()
    holder = nn.Module()
    holder.up_proj = gl
    holder, _ = inject_peft_and_freeze(holder, LoRAMethod(rank=2, alpha=2))
    assert isinstance(holder.up_proj, LoRAGroupedLinear)

    with torch.no_grad():
        holder.up_proj.lora_B.normal_()
    sizes = torch.tensor([3, 0, 2, 1])
    x = torch.randn(6, 8)
    with torch.no_grad():
        y = holder.up_proj(x, sizes)
    merged = LoRAMethod(rank=2, alpha=2).merge(holder)
    with torch.no_grad():
        y2 = merged.up_proj(x, sizes)
    torch.testing.assert_close(y, y2, rtol=1e-4, atol=1e-5)


def test_full_tune_and_stack():
    model = nn.Sequential()
    model.add_module("a_proj", nn.Linear(4, 4))
    model.add_module("head", nn.Linear(4, 2))
    model, _ = inject_peft_and_freeze(
        model, PeftStack(LoRAMethod(rank=2, alpha=2, target_patterns=(r".*a_proj$",)),
                         FullTuneMethod((r"head\..*",)))
    )
    assert isinstance(model.a_proj, LoRALinear)
    assert model.head.weight.requires_grad
    assert not model.a_proj.base.weight.requires_grad
