"""Distributed column standardization.

ref: nodes.stats.StandardScaler / StandardScalerModel.
Column mean/variance via per-rank partial sums + ONE fused RCCL all-reduce
(Spark: treeAggregate of column stats — SURVEY.md §2.10).
"""

from __future__ import annotations

import torch

from ...parallel.dataset import TensorDataset
from ...workflow.pipeline import BatchTransformer, Estimator


class StandardScalerModel(BatchTransformer):
    def __init__(self, mean: torch.Tensor, std: torch.Tensor | None = None):
        self.mean = mean
        self.std = std

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        self.mean = self.mean.to(device=X.device, dtype=X.dtype)
        if self.std is None:
            return X - self.mean
        self.std = self.std.to(device=X.device, dtype=X.dtype)
        return (X - self.mean) / self.std

    @classmethod
    def from_saved(cls, config, state):
        return cls(state["mean"], state.get("std"))


class StandardScaler(Estimator):
    def __init__(self, normalize_std_dev: bool = True, eps: float = 1e-12):
        self.normalize_std_dev = normalize_std_dev
        self.eps = eps

    def _fit(self, data: TensorDataset) -> StandardScalerModel:
        X = data.data
        rt = data.runtime
        n = torch.tensor([float(X.shape[0])], dtype=torch.float64, device=X.device)
        s1 = X.sum(0, dtype=torch.float64)
        stats = [n, s1]
        if self.normalize_std_dev:
            stats.append((X.to(torch.float64) ** 2).sum(0))
        rt.all_reduce_many_(stats)
        n_tot = stats[0].item()
        mean = (stats[1] / n_tot)
        std = None
        if self.normalize_std_dev:
            # unbiased variance, matching Spark/Breeze column std
            var = (stats[2] - n_tot * mean ** 2) / max(n_tot - 1, 1.0)
            std = var.clamp_min(self.eps).sqrt().to(X.dtype)
        return StandardScalerModel(mean.to(X.dtype), std)
