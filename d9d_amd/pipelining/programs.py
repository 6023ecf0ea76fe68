"""Schedule program builders (reference: d9d/pipelining/infra/schedule/program/).

Each builder returns, for one pp rank, a compute-only `Program`; the factory
wraps it with communication actions. Stage→rank topology is the LOOP style
(global stage g lives on rank g % pp, local index g // pp); the V style
(zig-zag) is used by ZBV/DualPipeV.
"""

from .actions import Action, ActionKind, Program


def loop_stage_to_rank(num_stages: int, pp: int) -> list[int]:
    return [g % pp for g in range(num_stages)]


def v_stage_to_rank(num_stages: int, pp: int) -> list[int]:
    """Zig-zag: ranks 0..pp-1 then pp-1..0 (requires num_stages = 2k*pp)."""
    out = []
    direction = 1
    r = 0
    for _ in range(num_stages):
        out.append(r)
        nr = r + direction
        if nr == pp:
            direction = -1
            nr = pp - 1
        elif nr < 0:
            direction = 1
            nr = 0
        r = nr
    return out


def local_stages(stage_to_rank: list[int], rank: int) -> list[int]:
    """Global stage ids owned by `rank`, in global order."""
    return [g for g, r in enumerate(stage_to_rank) if r == rank]


def build_gpipe(rank: int, pp: int, num_stages: int, num_microbatches: int,
                forward_only: bool = False) -> Program:
    """All forwards (per local stage in global order), then all backwards."""
    owned = local_stages(loop_stage_to_rank(num_stages, pp), rank)
    prog: Program = []
    for li, _ in enumerate(owned):
        for mb in range(num_microbatches):
            prog.append(Action(ActionKind.FORWARD_COMPUTE, li, mb))
    if not forward_only:
        for li in reversed(range(len(owned))):
            for mb in range(num_microbatches):
                prog.append(Action(ActionKind.BACKWARD_COMPUTE, li, mb))
    return prog


def build_looped_bfs(rank: int, pp: int, num_stages: int, num_microbatches: int,
                     forward_only: bool = False) -> Program:
    """BFS over stages: same as gpipe at program level (interleaving emerges
    from cross-rank overlap); reference: bfs.py:14."""
    return build_gpipe(rank, pp, num_stages, num_microbatches, forward_only)


def build_1f1b(rank: int, pp: int, num_stages: int, num_microbatches: int,
               zero_bubble: bool = False) -> Program:
    """Non-interleaved 1F1B (num_stages == pp). Warmup = pp-1-rank forwards,
    then steady 1F1B, then drain. zero_bubble: split B into B_in at B's slot
    and B_w deferred to the tail (ZB1P, reference: interleaved.py)."""
    if num_stages != pp:
        return build_interleaved_1f1b(rank, pp, num_stages, num_microbatches, zero_bubble)
    warmup = min(pp - 1 - rank, num_microbatches)
    prog: Program = []
    b_kind = ActionKind.BACKWARD_INPUT if zero_bubble else ActionKind.BACKWARD_COMPUTE
    fwd_mb = 0
    bwd_mb = 0
    for _ in range(warmup):
        prog.append(Action(ActionKind.FORWARD_COMPUTE, 0, fwd_mb))
        fwd_mb += 1
    # steady: 1F then 1B
    while fwd_mb < num_microbatches:
        prog.append(Action(ActionKind.FORWARD_COMPUTE, 0, fwd_mb))
        fwd_mb += 1
        prog.append(Action(b_kind, 0, bwd_mb))
        if zero_bubble:
            prog.append(Action(ActionKind.BACKWARD_WEIGHT, 0, bwd_mb))
        bwd_mb += 1
    while bwd_mb < num_microbatches:
        prog.append(Action(b_kind, 0, bwd_mb))
        if zero_bubble:
            prog.append(Action(ActionKind.BACKWARD_WEIGHT, 0, bwd_mb))
        bwd_mb += 1
    return prog


def build_zb1p(rank: int, pp: int, num_stages: int, num_microbatches: int) -> Program:
    """ZB1P: 1F1B with input/weight-split backward; weight grads deferred so
    d(input) propagates to the previous stage as early as possible."""
    assert num_stages == pp
    warmup = min(pp - 1 - rank, num_microbatches)
    prog: Program = []
    fwd_mb = 0
    bwd_mb = 0
    w_queue: list[int] = []
    for _ in range(warmup):
        prog.append(Action(ActionKind.FORWARD_COMPUTE, 0, fwd_mb))
        fwd_mb += 1
    while fwd_mb < num_microbatches:
        prog.append(Action(ActionKind.FORWARD_COMPUTE, 0, fwd_mb))
        fwd_mb += 1
        prog.append(Action(ActionKind.BACKWARD_INPUT, 0, bwd_mb))
        w_queue.append(bwd_mb)
        bwd_mb += 1
        # fill the bubble on late ranks: flush one deferred W per steady slot
        if rank >= pp - 1 and w_queue:
            prog.append(Action(ActionKind.BACKWARD_WEIGHT, 0, w_queue.pop(0)))
    while bwd_mb < num_microbatches:
        prog.append(Action(ActionKind.BACKWARD_INPUT, 0, bwd_mb))
        w_queue.append(bwd_mb)
        bwd_mb += 1
        if w_queue:
            prog.append(Action(ActionKind.BACKWARD_WEIGHT, 0, w_queue.pop(0)))
    for mb in w_queue:
        prog.append(Action(ActionKind.BACKWARD_WEIGHT, 0, mb))
    return prog


def build_zbv(rank: int, pp: int, num_stages: int, num_microbatches: int) -> Program:
    """Zero-bubble V (ZBV): V topology (rank r owns global stages r and
    2*pp-1-r), input/weight-split backward, weight grads deferred into the
    drain bubble (reference: zerobubblev.py; program order is BFS over the V
    legs -- a valid ZBV ordering, not the paper-optimal interleave).

    Local stage indices follow `local_stages(v_stage_to_rank(...), rank)`
    order (global ascending): 0 = down-leg stage (global r), 1 = up-leg
    stage (global 2*pp-1-r).
    """
    assert num_stages == 2 * pp, "ZBV requires exactly 2 stages per rank"
    prog: Program = []
    # forwards: down leg then up leg, all microbatches each
    for li in (0, 1):
        for mb in range(num_microbatches):
            prog.append(Action(ActionKind.FORWARD_COMPUTE, li, mb))
    # backwards: up leg then down leg, input-grad first, weights deferred
    w_queue: list = []
    for li in (1, 0):
        for mb in range(num_microbatches):
            prog.append(Action(ActionKind.BACKWARD_INPUT, li, mb))
            w_queue.append((li, mb))
            # fill the drain bubble with one deferred weight pass per slot
            if li == 0 and w_queue:
                wli, wmb = w_queue.pop(0)
                prog.append(Action(ActionKind.BACKWARD_WEIGHT, wli, wmb))
    for wli, wmb in w_queue:
        prog.append(Action(ActionKind.BACKWARD_WEIGHT, wli, wmb))
    return prog


def build_dualpipev(rank: int, pp: int, num_stages: int, num_microbatches: int) -> Program:
    """DualPipeV: bidirectional V schedule with interleaved forward/backward
    in the steady state and a zero-bubble weight-grad ramp in the drain
    (DeepSeek DualPipe, V variant; reference: dualpipev.py -- which also
    executes its F/B "compose" pairs sequentially at runtime, so a flattened
    F-then-B emission is behaviorally identical while the batched async P2P
    layer provides the communication overlap).

    Local stage 0 = down leg (global stage `rank`), 1 = up leg (global
    `2*pp-1-rank`). Requires num_microbatches >= 2*pp.
    """
    assert num_stages == 2 * pp, "DualPipeV requires exactly 2 stages per rank"
    assert num_microbatches >= num_stages, (
        f"DualPipeV requires num_microbatches ({num_microbatches}) >= "
        f"num_stages ({num_stages})"
    )
    prog: Program = []
    f = [0, 0]
    b = [0, 0]
    w_queue: list[tuple[int, int]] = []

    def add_f(s: int) -> None:
        prog.append(Action(ActionKind.FORWARD_COMPUTE, s, f[s]))
        f[s] += 1

    def add_b_full(s: int) -> None:
        prog.append(Action(ActionKind.BACKWARD_COMPUTE, s, b[s]))
        b[s] += 1

    def add_b_input(s: int) -> None:
        prog.append(Action(ActionKind.BACKWARD_INPUT, s, b[s]))
        w_queue.append((s, b[s]))
        b[s] += 1

    def pop_w() -> None:
        if w_queue:
            ws, wmb = w_queue.pop(0)
            prog.append(Action(ActionKind.BACKWARD_WEIGHT, ws, wmb))

    # 1: startup on the down leg
    for _ in range((pp - rank - 1) * 2):
        add_f(0)
    # 2: fill both legs
    for _ in range(rank + 1):
        add_f(0)
        add_f(1)
    # 3: up-leg input-grad / deferred-weight / forward mix
    for _ in range(pp - rank - 1):
        add_b_input(1)
        pop_w()
        add_f(1)
    # 4: steady state -- interleave F(down)/B(up) and F(up)/B(down)
    for _ in range(num_microbatches - 2 * pp + rank + 1):
        add_f(0)
        add_b_full(1)
        add_f(1)
        add_b_full(0)
    # 5: cooldown with remaining up-leg forwards
    for _ in range(pp - rank - 1):
        add_b_full(1)
        add_f(1)
        add_b_full(0)
    # 6: drain both legs, ramping into input-only backwards (zero bubble)
    enable_zb = False
    n6 = rank + 1
    for i in range(n6):
        if i == n6 // 2 and rank % 2 == 1:
            enable_zb = True
        (add_b_input if enable_zb else add_b_full)(1)
        if i == n6 // 2 and rank % 2 == 0:
            enable_zb = True
        (add_b_input if enable_zb else add_b_full)(0)
    # 7: deferred weights interleaved with down-leg input backwards
    for _ in range(pp - rank - 1):
        pop_w()
        add_b_input(0)
    # 8: flush remaining weight grads
    for _ in range(rank + 1):
        pop_w()
    return prog


def build_interleaved_1f1b(rank: int, pp: int, num_stages: int,
                           num_microbatches: int, zero_bubble: bool = False) -> Program:
    """Interleaved 1F1B (Megatron-style virtual pipeline; reference:
    interleaved.py Interleaved1F1BPipelineProgramBuilder). v = num_stages/pp
    chunks per rank on a LOOP topology (rank r owns global stages r, r+pp,
    ...; local chunk index == position in that list). Forward walks chunks in
    pp-sized rounds, backward walks them reversed; warmup depth
    (pp-1-rank)*2 + (v-1)*pp keeps every rank one chunk ahead.

    zero_bubble splits each backward into input-grad at its slot + deferred
    weight-grad at the tail."""
    assert num_stages % pp == 0
    v = num_stages // pp
    total = num_microbatches * v
    # the round-robin chunk walk needs whole pp-rounds; microbatch counts that
    # are not multiples of pp would interleave incorrectly
    assert num_microbatches % pp == 0 or v == 1, (
        "interleaved 1F1B needs num_microbatches % pp == 0"
    )

    def fwd_unit(i: int) -> tuple[int, int]:
        chunk = (i % (pp * v)) // pp
        round_base = (i // (pp * v)) * pp
        mb = round_base + i % pp
        return chunk, mb

    def bwd_unit(j: int) -> tuple[int, int]:
        chunk = v - 1 - (j % (pp * v)) // pp
        round_base = (j // (pp * v)) * pp
        mb = round_base + j % pp
        return chunk, mb

    b_kind = ActionKind.BACKWARD_INPUT if zero_bubble else ActionKind.BACKWARD_COMPUTE
    prog: Program = []
    w_queue: list[tuple[int, int]] = []

    def emit_f(i: int) -> None:
        chunk, mb = fwd_unit(i)
        prog.append(Action(ActionKind.FORWARD_COMPUTE, chunk, mb))

    def emit_b(j: int) -> None:
        chunk, mb = bwd_unit(j)
        prog.append(Action(b_kind, chunk, mb))
        if zero_bubble:
            w_queue.append((chunk, mb))

    warmup = min((pp - rank - 1) * 2 + (v - 1) * pp, total)
    f = b = 0
    for _ in range(warmup):
        emit_f(f)
        f += 1
    while f < total:
        emit_f(f)
        f += 1
        emit_b(b)
        b += 1
    while b < total:
        emit_b(b)
        b += 1
    for chunk, mb in w_queue:
        prog.append(Action(ActionKind.BACKWARD_WEIGHT, chunk, mb))
    return prog
