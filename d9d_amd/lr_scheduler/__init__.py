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
]
