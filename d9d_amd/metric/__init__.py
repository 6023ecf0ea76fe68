from .abc import Metric
from .accumulator import MetricAccumulator
from .impl import (
    WeightedMeanMetric,
    SumMetric,
    BinaryAUROC,
    ConfusionMatrixMetric,
    ComposeMetric,
)

__all__ = [
    "Metric",
    "MetricAccumulator",
    "WeightedMeanMetric",
    "SumMetric",
    "BinaryAUROC",
    "ConfusionMatrixMetric",
    "ComposeMetric",
]
