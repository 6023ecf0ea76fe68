"""PipelineStage: one model chunk + its P2P buffers and autograd bookkeeping.

Reference: d9d/pipelining/infra/stage/stage.py. Recv buffers are preallocated
from meta-device shape inference; forward caches (inputs, outputs) per
microbatch; backward supports full or input/weight-split (splitgrad).
"""

from typing import Any

import torch
from torch import nn

from ..core.autograd import GLOBAL_GRAD_CONTEXT, GradDirection


class PipelineStage(nn.Module):
    def __init__(
        self,
        module: nn.Module,
        stage_index: int,
        num_stages: int,
        device: torch.device,
    ) -> None:
        super().__init__()
        self.module = module
        self.stage_index = stage_index
        self.num_stages = num_stages
        self.device = device
        self.is_first = stage_index == 0
        self.is_last = stage_index == num_stages - 1

        self._recv_buffers: dict[int, dict[str, torch.Tensor]] = {}
        self._input_cache: dict[int, dict[str, torch.Tensor]] = {}
        self._output_cache: dict[int, dict[str, torch.Tensor]] = {}
        self._weight_backward_ctx: dict[int, tuple] = {}
        self._input_shapes: dict[str, tuple] = {}
        self._output_shapes: dict[str, tuple] = {}

    # -- buffers ---------------------------------------------------------------

    def configure_buffers(
        self, pipeline_inputs: dict[str, Any], num_microbatches: int
    ) -> None:
        self.reset()
        meta_inputs = {
            k: (v.to("meta") if isinstance(v, torch.Tensor) else v)
            for k, v in pipeline_inputs.items()
        }
        infer_in = getattr(self.module, "infer_stage_inputs_from_pipeline_inputs")
        infer_out = getattr(self.module, "infer_stage_outputs_from_pipeline_inputs")
        in_spec = infer_in(meta_inputs, num_microbatches)
        out_spec = infer_out(meta_inputs, num_microbatches)
        self._input_shapes = {k: (tuple(v.shape), v.dtype) for k, v in in_spec.items()}
        self._output_shapes = {k: (tuple(v.shape), v.dtype) for k, v in out_spec.items()}
        self._recv_buffers = {
            mb: {
                k: torch.empty(shape, dtype=dtype, device=self.device)
                for k, (shape, dtype) in self._input_shapes.items()
            }
            for mb in range(num_microbatches)
        }

    def reset(self) -> None:
        if self._input_cache or self._output_cache or self._weight_backward_ctx:
            stale = set(self._output_cache) | set(self._weight_backward_ctx)
            if stale:
                raise RuntimeError(
                    f"stage {self.stage_index}: dangling microbatch caches {sorted(stale)}"
                )
        self._input_cache.clear()
        self._output_cache.clear()
        self._weight_backward_ctx.clear()

    def recv_buffer(self, mb: int) -> dict[str, torch.Tensor]:
        return self._recv_buffers[mb]

    def grad_recv_buffer(self, mb: int) -> dict[str, torch.Tensor]:
        # gradient arrives with the shapes of this stage's OUTPUTS
        key = f"_grad_{mb}"
        if not hasattr(self, "_grad_buffers"):
            self._grad_buffers = {}
        if mb not in self._grad_buffers:
            self._grad_buffers[mb] = {
                k: torch.empty(shape, dtype=dtype, device=self.device)
                for k, (shape, dtype) in self._output_shapes.items()
            }
        return self._grad_buffers[mb]

    # -- compute ---------------------------------------------------------------

    def forward_one_chunk(
        self,
        mb: int,
        stage_inputs: dict[str, torch.Tensor],
        pipeline_kwargs: dict[str, Any],
    ) -> dict[str, torch.Tensor]:
        inputs = {}
        for k, v in stage_inputs.items():
            # Only ACTIVATION inputs (P2P tensors) participate in the stage
            # boundary autograd cut; extra pipeline inputs (labels, masks)
            # pass through untouched.
            if (
                isinstance(v, torch.Tensor)
                and v.is_floating_point()
                and not self.is_first
                and k in self._input_shapes
            ):
                v = v.detach().requires_grad_(True)
            inputs[k] = v
        outputs = self.module(**inputs, **pipeline_kwargs)
        self._input_cache[mb] = inputs
        self._output_cache[mb] = outputs
        return outputs

    def _collect_backward_edges(self, mb: int, output_grads: dict[str, torch.Tensor] | None,
                                loss: torch.Tensor | None):
        outputs = self._output_cache[mb]
        tensors: list[torch.Tensor] = []
        grads: list[torch.Tensor] = []
        if loss is not None:
            tensors.append(loss)
            grads.append(torch.ones_like(loss))
        else:
            assert output_grads is not None
            for k, g in output_grads.items():
                out = outputs[k]
                if out.requires_grad:
                    tensors.append(out)
                    grads.append(g)
        return tensors, grads

    def _stage_input_tensors(self, mb: int) -> list[torch.Tensor]:
        return [
            v
            for k, v in self._input_cache[mb].items()
            if k in self._input_shapes
            and isinstance(v, torch.Tensor) and v.requires_grad and v.is_leaf
        ]

    def backward_one_chunk(
        self,
        mb: int,
        output_grads: dict[str, torch.Tensor] | None = None,
        loss: torch.Tensor | None = None,
    ) -> dict[str, torch.Tensor]:
        """Full backward; returns grads for this stage's inputs (to send back)."""
        tensors, grads = self._collect_backward_edges(mb, output_grads, loss)
        torch.autograd.backward(tensors, grads)
        input_grads = {
            k: v.grad
            for k, v in self._input_cache[mb].items()
            if k in self._input_shapes
            and isinstance(v, torch.Tensor) and v.requires_grad and v.is_leaf
        }
        del self._input_cache[mb]
        del self._output_cache[mb]
        return input_grads

    def backward_input_only(
        self,
        mb: int,
        output_grads: dict[str, torch.Tensor] | None = None,
        loss: torch.Tensor | None = None,
    ) -> dict[str, torch.Tensor]:
        """Zero-bubble phase 1: d(inputs) only; weight grads deferred."""
        tensors, grads = self._collect_backward_edges(mb, output_grads, loss)
        stage_inputs = self._stage_input_tensors(mb)
        if stage_inputs:
            with GLOBAL_GRAD_CONTEXT.with_directions(GradDirection.INPUTS):
                torch.autograd.backward(
                    tensors, grads, inputs=stage_inputs, retain_graph=True
                )
        self._weight_backward_ctx[mb] = (tensors, grads)
        input_grads = {
            k: v.grad
            for k, v in self._input_cache[mb].items()
            if k in self._input_shapes
            and isinstance(v, torch.Tensor) and v.requires_grad and v.is_leaf
        }
        return input_grads

    def backward_weight_only(self, mb: int) -> None:
        """Zero-bubble phase 2: d(weights) from the retained graph."""
        tensors, grads = self._weight_backward_ctx.pop(mb)
        params = [p for p in self.module.parameters() if p.requires_grad]
        if params:
            with GLOBAL_GRAD_CONTEXT.with_directions(GradDirection.WEIGHTS):
                torch.autograd.backward(tensors, grads, inputs=params)
        del self._input_cache[mb]
        del self._output_cache[mb]
