"""Pipeline action taxonomy (reference: d9d/pipelining/infra/schedule/component/runtime/action.py).

A *program* is a per-pp-rank list of actions over (local stage index,
microbatch). Compute actions short-circuit P2P when the peer stage lives on
the same rank.
"""

import enum
from dataclasses import dataclass


class ActionKind(enum.Enum):
    FORWARD_RECV = "F_recv"
    FORWARD_COMPUTE = "F"
    FORWARD_SEND = "F_send"
    BACKWARD_RECV = "B_recv"
    BACKWARD_COMPUTE = "B"        # full backward (inputs + weights)
    BACKWARD_INPUT = "B_in"       # zero-bubble: d(input) only
    BACKWARD_WEIGHT = "B_w"       # zero-bubble: d(weights), deferred
    BACKWARD_SEND = "B_send"


@dataclass(frozen=True)
class Action:
    kind: ActionKind
    stage: int       # LOCAL stage index on this rank
    microbatch: int

    def __str__(self) -> str:
        return f"{self.kind.value}(s{self.stage},mb{self.microbatch})"


Program = list[Action]


def insert_communication_actions(
    compute_program: Program,
    stage_is_first: list[bool],
    stage_is_last: list[bool],
    prev_is_local: list[bool],
    next_is_local: list[bool],
) -> Program:
    """Wrap a compute-only program with send/recv actions.

    For stage s of this rank: a FORWARD_COMPUTE consuming remote activations
    gets a FORWARD_RECV before it and a FORWARD_SEND after it when the
    consumer is remote; symmetric for backward. NCCL P2P has no tags, so the
    per-(src,dst) op order is the program order — builders emit microbatches
    in the same order on both sides by construction.
    """
    out: Program = []
    for action in compute_program:
        s = action.stage
        if action.kind is ActionKind.FORWARD_COMPUTE:
            if not stage_is_first[s] and not prev_is_local[s]:
                out.append(Action(ActionKind.FORWARD_RECV, s, action.microbatch))
            out.append(action)
            if not stage_is_last[s] and not next_is_local[s]:
                out.append(Action(ActionKind.FORWARD_SEND, s, action.microbatch))
        elif action.kind in (ActionKind.BACKWARD_COMPUTE, ActionKind.BACKWARD_INPUT):
            if not stage_is_last[s] and not next_is_local[s]:
                out.append(Action(ActionKind.BACKWARD_RECV, s, action.microbatch))
            out.append(action)
            if not stage_is_first[s] and not prev_is_local[s]:
                out.append(Action(ActionKind.BACKWARD_SEND, s, action.microbatch))
        else:
            out.append(action)
    return out
