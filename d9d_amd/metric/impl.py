"""Metric implementations (reference: d9d/metric/impl/)."""

from typing import Any

import torch

from .abc import Metric
from .accumulator import MetricAccumulator


class WeightedMeanMetric(Metric):
    def __init__(self) -> None:
        self.acc = MetricAccumulator()
        self.weight = MetricAccumulator()

    def update(self, value, weight=1.0) -> None:
        v = torch.as_tensor(value, dtype=torch.float64)
        w = torch.as_tensor(weight, dtype=torch.float64)
        self.acc.add_((v * w).cpu() if v.is_cuda else v * w)
        self.weight.add_(w.cpu() if w.is_cuda else w)

    def sync(self, group=None) -> None:
        self.acc.sync(group)
        self.weight.sync(group)

    def compute(self) -> float:
        w = self.weight.synced.item()
        return self.acc.synced.item() / w if w else 0.0

    def reset(self) -> None:
        self.acc.reset()
        self.weight.reset()

    def to(self, device):
        return self  # accumulates on host

    def state_dict(self):
        return {"acc": self.acc.local, "weight": self.weight.local}

    def load_state_dict(self, sd):
        self.acc.local = sd["acc"]
        self.weight.local = sd["weight"]


class SumMetric(Metric):
    def __init__(self) -> None:
        self.acc = MetricAccumulator()

    def update(self, value) -> None:
        v = torch.as_tensor(value, dtype=torch.float64)
        self.acc.add_(v.cpu() if v.is_cuda else v)

    def sync(self, group=None) -> None:
        self.acc.sync(group)

    def compute(self) -> float:
        return self.acc.synced.item()

    def reset(self) -> None:
        self.acc.reset()

    def to(self, device):
        return self

    def state_dict(self):
        return {"acc": self.acc.local}

    def load_state_dict(self, sd):
        self.acc.local = sd["acc"]


class BinaryAUROC(Metric):
    """Histogram-based AUROC (reference: impl/classification/auroc.py:48)."""

    def __init__(self, num_bins: int = 256) -> None:
        self.num_bins = num_bins
        self.pos = MetricAccumulator((num_bins,))
        self.neg = MetricAccumulator((num_bins,))

    def update(self, scores: torch.Tensor, targets: torch.Tensor) -> None:
        scores = scores.detach().float().clamp(0, 1).cpu()
        targets = targets.detach().bool().cpu()
        bins = (scores * (self.num_bins - 1)).long()
        pos_hist = torch.bincount(bins[targets], minlength=self.num_bins)
        neg_hist = torch.bincount(bins[~targets], minlength=self.num_bins)
        self.pos.add_(pos_hist.double())
        self.neg.add_(neg_hist.double())

    def sync(self, group=None) -> None:
        self.pos.sync(group)
        self.neg.sync(group)

    def compute(self) -> float:
        pos = self.pos.synced
        neg = self.neg.synced
        total_pos = pos.sum()
        total_neg = neg.sum()
        if total_pos == 0 or total_neg == 0:
            return 0.5
        # Sweep thresholds from high to low scores.
        tps = torch.flip(torch.cumsum(torch.flip(pos, [0]), 0), [0])
        fps = torch.flip(torch.cumsum(torch.flip(neg, [0]), 0), [0])
        tpr = torch.cat([torch.zeros(1, dtype=torch.float64), torch.flip(tps, [0]) / total_pos])
        fpr = torch.cat([torch.zeros(1, dtype=torch.float64), torch.flip(fps, [0]) / total_neg])
        return torch.trapz(tpr, fpr).item()

    def reset(self) -> None:
        self.pos.reset()
        self.neg.reset()

    def to(self, device):
        return self

    def state_dict(self):
        return {"pos": self.pos.local, "neg": self.neg.local}

    def load_state_dict(self, sd):
        self.pos.local = sd["pos"]
        self.neg.local = sd["neg"]


class ConfusionMatrixMetric(Metric):
    """Multiclass confusion matrix + micro/macro/weighted P/R/F1
    (reference: impl/classification/confusion_matrix.py:23,105)."""

    def __init__(self, num_classes: int, average: str = "macro") -> None:
        assert average in ("micro", "macro", "weighted")
        self.num_classes = num_classes
        self.average = average
        self.matrix = MetricAccumulator((num_classes, num_classes))

    def update(self, preds: torch.Tensor, targets: torch.Tensor) -> None:
        preds = preds.detach().reshape(-1).long().cpu()
        targets = targets.detach().reshape(-1).long().cpu()
        idx = targets * self.num_classes + preds
        counts = torch.bincount(idx, minlength=self.num_classes**2)
        self.matrix.add_(counts.reshape(self.num_classes, self.num_classes).double())

    def sync(self, group=None) -> None:
        self.matrix.sync(group)

    def compute(self) -> dict[str, Any]:
        m = self.matrix.synced
        tp = m.diagonal()
        support = m.sum(dim=1)
        predicted = m.sum(dim=0)
        if self.average == "micro":
            precision = recall = tp.sum() / m.sum().clamp_min(1)
            f1 = precision
            return {
                "precision": precision.item(),
                "recall": recall.item(),
                "f1": f1.item(),
                "accuracy": (tp.sum() / m.sum().clamp_min(1)).item(),
            }
        per_p = tp / predicted.clamp_min(1)
        per_r = tp / support.clamp_min(1)
        per_f1 = 2 * per_p * per_r / (per_p + per_r).clamp_min(1e-12)
        if self.average == "macro":
            w = torch.ones_like(support) / self.num_classes
        else:
            w = support / support.sum().clamp_min(1)
        return {
            "precision": (per_p * w).sum().item(),
            "recall": (per_r * w).sum().item(),
            "f1": (per_f1 * w).sum().item(),
            "accuracy": (tp.sum() / m.sum().clamp_min(1)).item(),
        }

    def reset(self) -> None:
        self.matrix.reset()

    def to(self, device):
        return self

    def state_dict(self):
        return {"matrix": self.matrix.local}

    def load_state_dict(self, sd):
        self.matrix.local = sd["matrix"]


class ComposeMetric(Metric):
    """Named bundle of metrics (reference: impl/container/compose.py:10)."""

    def __init__(self, **metrics: Metric) -> None:
        self.metrics = metrics

    def update(self, name: str, *args, **kwargs) -> None:
        self.metrics[name].update(*args, **kwargs)

    def sync(self, group=None) -> None:
        for m in self.metrics.values():
            m.sync(group)

    def compute(self) -> dict[str, Any]:
        return {name: m.compute() for name, m in self.metrics.items()}

    def reset(self) -> None:
        for m in self.metrics.values():
            m.reset()

    def to(self, device):
        for m in self.metrics.values():
            m.to(device)
        return self

    def state_dict(self):
        return {name: m.state_dict() for name, m in self.metrics.items()}

    def load_state_dict(self, sd):
        for name, m in self.metrics.items():
            if name in sd:
                m.load_state_dict(sd[name])
