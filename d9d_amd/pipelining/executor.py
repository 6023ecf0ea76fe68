"""Pipeline schedule executor (reference: runtime/executor.py + offline.py).

Interprets a per-rank action program over local PipelineStages, sharding the
step's inputs into microbatches, moving activations/gradients over RCCL P2P
(or local handoff when the adjacent stage lives on this rank), and invoking
the loss callback on last-stage microbatches.
"""

from typing import Any, Callable

import torch
from torch.profiler import record_function

from ..core.sharding import SpecShard, shard_tree
from .actions import Action, ActionKind, Program
from .comms import PipelineCommunicationHandler
from .stage import PipelineStage

# loss_fn(mb_index, outputs, mb_inputs) -> loss tensor (scalar) or None
PipelineLossFn = Callable[[int, dict[str, torch.Tensor], dict[str, Any]], torch.Tensor | None]


def _extra_stage_inputs(stage: PipelineStage, mb_inputs: dict, have: set) -> dict:
    """Pipeline inputs a non-first stage consumes directly.

    Modules may declare `pipeline_input_names()` (e.g. {"position_ids",
    "labels"}); the fallback passes only non-tensor extras.
    """
    names_fn = getattr(stage.module, "pipeline_input_names", None)
    if names_fn is not None:
        names = names_fn()
        return {k: v for k, v in mb_inputs.items() if k in names and k not in have}
    return {
        k: v
        for k, v in mb_inputs.items()
        if k not in have and not isinstance(v, torch.Tensor)
    }


class PipelineScheduleExecutor:
    def __init__(
        self,
        stages: list[PipelineStage],              # local stages, global order
        global_stage_ids: list[int],
        rank_of_stage: list[int],
        program: Program,
        num_microbatches: int,
        pp_rank: int,
        group=None,
        input_spec=None,
        forward_only: bool = False,
    ) -> None:
        self.forward_only = forward_only
        self.stages = stages
        self.global_stage_ids = global_stage_ids
        self.rank_of_stage = rank_of_stage
        self.program = program
        self.num_microbatches = num_microbatches
        self.pp_rank = pp_rank
        self.comm = PipelineCommunicationHandler(group, rank_of_stage)
        self.input_spec = input_spec or SpecShard(dim=0)
        self._local_of_global = {g: i for i, g in enumerate(global_stage_ids)}

    @property
    def has_first_stage(self) -> bool:
        return 0 in self.global_stage_ids

    @property
    def has_last_stage(self) -> bool:
        return (len(self.rank_of_stage) - 1) in self.global_stage_ids

    def configure_buffers(self, pipeline_inputs: dict[str, Any]) -> None:
        for stage in self.stages:
            stage.configure_buffers(pipeline_inputs, self.num_microbatches)

    def step(
        self,
        pipeline_inputs: dict[str, Any],
        pipeline_kwargs: dict[str, Any] | None = None,
        loss_fn: PipelineLossFn | None = None,
    ) -> list[torch.Tensor]:
        pipeline_kwargs = pipeline_kwargs or {}
        microbatches = shard_tree(pipeline_inputs, self.input_spec, self.num_microbatches)
        losses: list[torch.Tensor] = []
        # transient per-(stage, mb) stashes
        fwd_sendables: dict[tuple, dict] = {}
        bwd_sendables: dict[tuple, dict] = {}
        loss_cache: dict[tuple, torch.Tensor] = {}

        for action in self.program:
            with record_function(str(action)):
                self._run_action(
                    action, microbatches, pipeline_kwargs, loss_fn,
                    fwd_sendables, bwd_sendables, loss_cache, losses,
                )
        self.comm.wait_all()
        for stage in self.stages:
            stage.reset()
        return losses

    # -- helpers ---------------------------------------------------------------

    def _stage(self, local_idx: int) -> PipelineStage:
        return self.stages[local_idx]

    def _prev_info(self, local_idx: int):
        g = self.global_stage_ids[local_idx]
        if g == 0:
            return None
        prev_rank = self.rank_of_stage[g - 1]
        return (g - 1, prev_rank, prev_rank == self.pp_rank)

    def _next_info(self, local_idx: int):
        g = self.global_stage_ids[local_idx]
        if g == len(self.rank_of_stage) - 1:
            return None
        next_rank = self.rank_of_stage[g + 1]
        return (g + 1, next_rank, next_rank == self.pp_rank)

    def _run_action(
        self, action: Action, microbatches, pipeline_kwargs, loss_fn,
        fwd_sendables, bwd_sendables, loss_cache, losses,
    ) -> None:
        s, mb = action.stage, action.microbatch
        stage = self._stage(s)
        kind = action.kind

        if kind is ActionKind.FORWARD_RECV:
            prev = self._prev_info(s)
            self.comm.queue_recv(stage.recv_buffer(mb), prev[1], ("f", s, mb))

        elif kind is ActionKind.FORWARD_COMPUTE:
            if stage.is_first:
                names_fn = getattr(stage.module, "pipeline_input_names", None)
                if names_fn is not None:
                    names = names_fn()
                    stage_inputs = {
                        k: v for k, v in microbatches[mb].items() if k in names
                    }
                else:
                    stage_inputs = dict(microbatches[mb])
            else:
                prev = self._prev_info(s)
                if prev[2]:  # local handoff
                    prev_local = self._local_of_global[prev[0]]
                    outs = fwd_sendables.pop(("local_f", prev_local, mb))
                    stage_inputs = outs
                else:
                    self.comm.wait_recv(("f", s, mb))
                    stage_inputs = dict(stage.recv_buffer(mb))
                # later stages also consume their declared pipeline inputs
                # (e.g. position_ids everywhere, labels on the last stage)
                stage_inputs.update(
                    _extra_stage_inputs(stage, microbatches[mb], set(stage_inputs))
                )
            outputs = stage.forward_one_chunk(mb, stage_inputs, pipeline_kwargs)
            nxt = self._next_info(s)
            if nxt is not None:
                payload = {
                    k: outputs[k] for k in stage._output_shapes
                } if stage._output_shapes else dict(outputs)
                if nxt[2]:
                    next_local = self._local_of_global[nxt[0]]
                    fwd_sendables[("local_f", s, mb)] = payload
                else:
                    fwd_sendables[("remote_f", s, mb)] = payload
            if stage.is_last and loss_fn is not None:
                loss = loss_fn(mb, outputs, microbatches[mb])
                if loss is not None:
                    loss_cache[(s, mb)] = loss
                    losses.append(loss.detach())
            if self.forward_only:
                stage._input_cache.pop(mb, None)
                stage._output_cache.pop(mb, None)

        elif kind is ActionKind.FORWARD_SEND:
            nxt = self._next_info(s)
            payload = fwd_sendables.pop(("remote_f", s, mb))
            self.comm.queue_send(payload, nxt[1])

        elif kind is ActionKind.BACKWARD_RECV:
            nxt = self._next_info(s)
            self.comm.queue_recv(stage.grad_recv_buffer(mb), nxt[1], ("b", s, mb))

        elif kind in (ActionKind.BACKWARD_COMPUTE, ActionKind.BACKWARD_INPUT):
            output_grads = None
            loss = None
            if stage.is_last:
                loss = loss_cache.pop((s, mb), None)
                if loss is None:
                    raise RuntimeError(f"no cached loss for stage {s} mb {mb}")
            else:
                nxt = self._next_info(s)
                if nxt[2]:
                    next_local = self._local_of_global[nxt[0]]
                    output_grads = bwd_sendables.pop(("local_b", next_local, mb))
                else:
                    self.comm.wait_recv(("b", s, mb))
                    output_grads = dict(stage.grad_recv_buffer(mb))
            if kind is ActionKind.BACKWARD_COMPUTE:
                input_grads = stage.backward_one_chunk(mb, output_grads, loss)
            else:
                input_grads = stage.backward_input_only(mb, output_grads, loss)
            prev = self._prev_info(s)
            if prev is not None and input_grads:
                if prev[2]:
                    bwd_sendables[("local_b", s, mb)] = input_grads
                else:
                    bwd_sendables[("remote_b", s, mb)] = input_grads

        elif kind is ActionKind.BACKWARD_WEIGHT:
            stage.backward_weight_only(mb)

        elif kind is ActionKind.BACKWARD_SEND:
            prev = self._prev_info(s)
            payload = bwd_sendables.pop(("remote_b", s, mb))
            self.comm.queue_send(payload, prev[1])

        else:
            raise AssertionError(kind)


class OfflinePipelineExecutor:
    """Single-process fallback: runs all stages sequentially (reference: offline.py)."""

    def __init__(self, stages: list[PipelineStage], num_microbatches: int,
                 input_spec=None, forward_only: bool = False) -> None:
        self.stages = stages
        self.num_microbatches = num_microbatches
        self.input_spec = input_spec or SpecShard(dim=0)
        self.has_first_stage = True
        self.has_last_stage = True
        self.forward_only = forward_only

    def configure_buffers(self, pipeline_inputs: dict[str, Any]) -> None:
        for stage in self.stages:
            stage.configure_buffers(pipeline_inputs, self.num_microbatches)

    def step(self, pipeline_inputs, pipeline_kwargs=None, loss_fn=None):
        pipeline_kwargs = pipeline_kwargs or {}
        microbatches = shard_tree(pipeline_inputs, self.input_spec, self.num_microbatches)
        losses = []
        for mb in range(self.num_microbatches):
            outs = None
            carried: dict[str, Any] = {}
            for stage in self.stages:
                if stage.is_first:
                    names_fn = getattr(stage.module, "pipeline_input_names", None)
                    if names_fn is not None:
                        names = names_fn()
                        carried = {
                            k: v for k, v in microbatches[mb].items() if k in names
                        }
                    else:
                        carried = dict(microbatches[mb])
                else:
                    carried = {k: outs[k] for k in stage._input_shapes} if stage._input_shapes else dict(outs)
                    carried.update(
                        _extra_stage_inputs(stage, microbatches[mb], set(carried))
                    )
                outs = stage.forward_one_chunk(mb, carried, pipeline_kwargs)
            loss = loss_fn(mb, outs, microbatches[mb]) if loss_fn else None
            if loss is not None:
                losses.append(loss.detach())
            if self.forward_only:
                for stage in self.stages:
                    stage._input_cache.pop(mb, None)
                    stage._output_cache.pop(mb, None)
                continue
            # backward through all stages via autograd chain
            grads = None
            for si in reversed(range(len(self.stages))):
                stage = self.stages[si]
                if stage.is_last:
                    input_grads = stage.backward_one_chunk(mb, None, loss)
                else:
                    input_grads = stage.backward_one_chunk(mb, grads, None)
                grads = input_grads
        for stage in self.stages:
            stage.reset()
        return losses
