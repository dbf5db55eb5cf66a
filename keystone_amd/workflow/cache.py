"""Materialization nodes: Cacher (HBM) and Checkpointer (NVMe).

ref: workflow Cacher / Checkpointer.  On Spark, ``Cacher``
called ``rdd.cache()``; here "cache" means *keep the shard tensor resident
in HBM* — which datasets already are — so Cacher's real job is to act as a
materialization barrier the optimizer can splice in (AutoCacheRule) so that
a shared intermediate is computed once instead of once per consumer.
``Checkpointer`` additionally spills the shard to disk (per-rank file) and
reloads lazily, trading HBM for NVMe when the 288 GB budget is exceeded.
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from ..parallel.dataset import Dataset, ItemDataset, TensorDataset
from .pipeline import Transformer

__all__ = ["Cacher", "Checkpointer"]


class Cacher(Transformer):
    """Identity with materialization semantics (ref: nodes.util.Cacher)."""

    def __init__(self, name: Optional[str] = None):
        self.name = name

    def apply(self, x):
        return x

    def apply_dataset(self, ds: Dataset) -> Dataset:
        # Tensors are HBM-resident already; for lazy item datasets force the
        # list. The executor memoizes this node's output, so every consumer
        # shares the materialized value.
        if isinstance(ds, ItemDataset):
            ds._items = list(ds._items)
        return ds


class Checkpointer(Transformer):
    """Spill a TensorDataset shard to disk; identity otherwise
    (ref: workflow.Checkpointer)."""

    def __init__(self, path: str):
        self.path = path

    def apply(self, x):
        return x

    def apply_dataset(self, ds: Dataset) -> Dataset:
        if isinstance(ds, TensorDataset):
            os.makedirs(self.path, exist_ok=True)
            f = os.path.join(self.path, f"shard_{ds.runtime.rank}.pt")
            torch.save(ds.data.cpu(), f)
        return ds

    @staticmethod
    def load(path: str, runtime=None) -> TensorDataset:
        from ..parallel.runtime import get_runtime
        rt = runtime or get_runtime()
        f = os.path.join(path, f"shard_{rt.rank}.pt")
        return TensorDataset(torch.load(f).to(rt.device), rt)
