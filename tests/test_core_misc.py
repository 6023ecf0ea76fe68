import torch

from d9d_amd.core.autograd import GLOBAL_GRAD_CONTEXT, GradDirection
from d9d_amd.core.offload import offload_tensor, onload_tensor


def test_grad_context_defaults_to_both():
    assert GLOBAL_GRAD_CONTEXT.computes(GradDirection.INPUTS)
    assert GLOBAL_GRAD_CONTEXT.computes(GradDirection.WEIGHTS)


def test_grad_context_with_directions():
    with GLOBAL_GRAD_CONTEXT.with_directions(GradDirection.INPUTS):
        assert GLOBAL_GRAD_CONTEXT.computes(GradDirection.INPUTS)
        assert not GLOBAL_GRAD_CONTEXT.computes(GradDirection.WEIGHTS)
        with GLOBAL_GRAD_CONTEXT.with_directions(GradDirection.WEIGHTS):
            assert not GLOBAL_GRAD_CONTEXT.computes(GradDirection.INPUTS)
            assert GLOBAL_GRAD_CONTEXT.computes(GradDirection.WEIGHTS)
        assert GLOBAL_GRAD_CONTEXT.computes(GradDirection.INPUTS)
    assert GLOBAL_GRAD_CONTEXT.computes(GradDirection.WEIGHTS)


def test_offload_tensor_identity_preserved_cpu():
    t = torch.randn(4, 4)
    original_id = id(t)
    offload_tensor(t)  # cpu->cpu is a no-op
    assert id(t) == original_id
    onload_tensor(t, torch.device("cpu"))
    assert t.device.type == "cpu"
