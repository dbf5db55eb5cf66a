"""Classifier evaluators.

ref: evaluation.MulticlassClassifierEvaluator (confusion
matrix, total/per-class accuracy, macro F1) and BinaryClassifierEvaluator
(accuracy/precision/recall/F1).  The confusion matrix is accumulated per
rank and all-reduced (Spark: reduce of per-partition matrices).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

from ..parallel.dataset import Dataset, TensorDataset
from ..parallel.runtime import get_runtime


def _as_label_tensor(x) -> torch.Tensor:
    if isinstance(x, TensorDataset):
        return x.data
    if isinstance(x, Dataset):
        return torch.stack([torch.as_tensor(v) for v in x.items()])
    if hasattr(x, "get"):
        return _as_label_tensor(x.get())
    return torch.as_tensor(x)


@dataclass
class MulticlassMetrics:
    confusion: torch.Tensor  # [K, K] rows = actual, cols = predicted

    @property
    def total_accuracy(self) -> float:
        return float(self.confusion.diagonal().sum() / self.confusion.sum().clamp_min(1))

    @property
    def total_error(self) -> float:
        return 1.0 - self.total_accuracy

    @property
    def per_class_accuracy(self) -> torch.Tensor:
        return self.confusion.diagonal() / self.confusion.sum(1).clamp_min(1)

    @property
    def macro_f1(self) -> float:
        tp = self.confusion.diagonal()
        prec = tp / self.confusion.sum(0).clamp_min(1)
        rec = tp / self.confusion.sum(1).clamp_min(1)
        f1 = 2 * prec * rec / (prec + rec).clamp_min(1e-12)
        return float(f1.mean())

    def summary(self) -> str:
        return (f"accuracy={self.total_accuracy:.4f} "
                f"macroF1={self.macro_f1:.4f}")


class MulticlassClassifierEvaluator:
    def __init__(self, num_classes: int):
        self.num_classes = num_classes

    def evaluate(self, predictions, actuals) -> MulticlassMetrics:
        pred = _as_label_tensor(predictions).long().reshape(-1)
        act = _as_label_tensor(actuals).long().reshape(-1)
        K = self.num_classes
        conf = torch.zeros(K, K, dtype=torch.float64, device=pred.device)
        idx = act * K + pred
        conf.reshape(-1).index_add_(0, idx, torch.ones_like(idx, dtype=torch.float64))
        get_runtime().all_reduce_(conf)
        return MulticlassMetrics(conf.cpu())


@dataclass
class BinaryClassificationMetrics:
    tp: float
    fp: float
    tn: float
    fn: float

    @property
    def accuracy(self):
        t = self.tp + self.fp + self.tn + self.fn
        return (self.tp + self.tn) / max(t, 1)

    @property
    def precision(self):
        return self.tp / max(self.tp + self.fp, 1)

    @property
    def recall(self):
        return self.tp / max(self.tp + self.fn, 1)

    @property
    def f1(self):
        p, r = self.precision, self.recall
        return 2 * p * r / max(p + r, 1e-12)


class BinaryClassifierEvaluator:
    def evaluate(self, predictions, actuals) -> BinaryClassificationMetrics:
        pred = _as_label_tensor(predictions).reshape(-1) > 0.5
        act = _as_label_tensor(actuals).reshape(-1) > 0.5
        stats = torch.tensor([
            float((pred & act).sum()), float((pred & ~act).sum()),
            float((~pred & ~act).sum()), float((~pred & act).sum())],
            dtype=torch.float64)
        get_runtime().all_reduce_(stats)
        return BinaryClassificationMetrics(*[float(x) for x in stats])
