"""Remaining util nodes.

ref: nodes.util.Shuffler (randomly redistribute rows
across the cluster — Spark repartition/shuffle) and the WeightedNode
marker trait (per-node cost weight for the profiler, §2.1).
"""

from __future__ import annotations

import torch

from ...parallel.dataset import Dataset, TensorDataset
from ...workflow.pipeline import Transformer


class WeightedNode:
    """Marker mixin: expose a cost multiplier (e.g. solver epochs) so the
    Profiler can scale a sampled measurement to the real run
    (ref: WeightedNode / WeightedOperator)."""

    @property
    def weight(self) -> float:
        return float(getattr(self, "num_iter", 1) or 1)


class Shuffler(Transformer):
    """Deterministic pseudo-random redistribution of rows across ranks
    (ref: nodes.util.Shuffler; Spark shuffle -> RCCL all-to-all)."""

    def __init__(self, seed: int = 0):
        self.seed = seed

    def apply(self, x):
        return x

    def apply_dataset(self, ds: Dataset) -> Dataset:
        if not isinstance(ds, TensorDataset):
            return ds
        rt = ds.runtime
        X = ds.data
        g = torch.Generator().manual_seed(self.seed + 31 * rt.rank)
        perm = torch.randperm(X.shape[0], generator=g).to(X.device)
        X = X[perm]
        if not rt.is_distributed:
            return TensorDataset(X, rt)
        dest = torch.randint(0, rt.world_size, (X.shape[0],), generator=g)
        send = [X[(dest == r).to(X.device)] for r in range(rt.world_size)]
        recv = rt.all_to_all(send)
        return TensorDataset(torch.cat(recv, dim=0), rt)
