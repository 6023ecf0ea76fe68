from .visualizer import (
    render_lr_ascii,
    simulate_lr_history,
    visualize_lr_scheduler,
)
from .piecewise import (
    Curve,
    ConstantCurve,
    LinearCurve,
    CosineCurve,
    PolynomialCurve,
    ExponentialCurve,
    Phase,
    PiecewiseLRScheduler,
    piecewise_schedule,
)

__all__ = [
    "Curve",
    "ConstantCurve",
    "LinearCurve",
    "CosineCurve",
    "PolynomialCurve",
    "ExponentialCurve",
    "Phase",
    "PiecewiseLRScheduler",
    "piecewise_schedule",
    "simulate_lr_history",
    "render_lr_ascii",
    "visualize_lr_scheduler",
]
