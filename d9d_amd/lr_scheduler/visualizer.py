"""LR schedule visualizer (reference: d9d/lr_scheduler/visualizer.py).

The reference renders the simulated schedule with Plotly; this environment
has no plotting stack, so the MI355X-native equivalent simulates the
schedule the same way and renders a terminal/unicode sparkline plot (and can
dump the raw curve as CSV for external plotting).
"""

from collections.abc import Callable

from torch import nn
from torch.optim import SGD, Optimizer
from torch.optim.lr_scheduler import LRScheduler

SchedulerFactory = Callable[[Optimizer], LRScheduler]

_BARS = " ▁▂▃▄▅▆▇█"


def simulate_lr_history(
    factory: SchedulerFactory, num_steps: int, init_lr: float = 1.0
) -> list[float]:
    """Run the schedule against a dummy optimizer and record the LR per step."""
    optimizer = SGD(nn.Linear(1, 1).parameters(), lr=init_lr)
    scheduler = factory(optimizer)
    lrs = []
    for _ in range(num_steps):
        lrs.append(optimizer.param_groups[0]["lr"])
        scheduler.step()
    return lrs


def render_lr_ascii(lrs: list[float], width: int = 100, height: int = 1) -> str:
    """Compress the LR curve into a unicode sparkline (max-pooled per column)."""
    if not lrs:
        return ""
    lo, hi = min(lrs), max(lrs)
    span = (hi - lo) or 1.0
    cols = min(width, len(lrs))
    per = len(lrs) / cols
    pooled = [
        max(lrs[int(i * per) : max(int((i + 1) * per), int(i * per) + 1)])
        for i in range(cols)
    ]
    line = "".join(
        _BARS[min(int((v - lo) / span * (len(_BARS) - 1)), len(_BARS) - 1)]
        for v in pooled
    )
    return f"lr [{lo:.3e} .. {hi:.3e}] over {len(lrs)} steps\n{line}"


def visualize_lr_scheduler(
    factory: SchedulerFactory,
    num_steps: int,
    init_lr: float = 1.0,
    csv_path: str | None = None,
) -> str:
    """Simulate `num_steps` of the schedule and print a terminal plot.

    Returns the rendered string; optionally writes `step,lr` rows to
    `csv_path` for external plotting.
    """
    lrs = simulate_lr_history(factory, num_steps, init_lr)
    if csv_path is not None:
        with open(csv_path, "w") as f:
            f.write("step,lr\n")
            for i, v in enumerate(lrs):
                f.write(f"{i},{v}\n")
    rendered = render_lr_ascii(lrs)
    print(rendered)
    return rendered
