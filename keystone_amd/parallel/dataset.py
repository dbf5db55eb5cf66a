"""Partitioned datasets: the RDD replacement.

KeystoneML pipelines are lifted over Spark ``RDD[T]`` (ref: workflow package,
SURVEY.md L0/L2).  The MI355X rebuild keeps *one shard per rank resident in
HBM* instead of many small partitions over a cluster:

  * ``Dataset``        — abstract distributed handle (local shard + runtime).
  * ``TensorDataset``  — the common case: the local shard is ONE 2-D (or N-D)
                         torch tensor, rows = local examples.  Batched matrix
                         ops (GEMM featurizers, solvers) run on it directly —
                         this is what Spark's ``mapPartitions`` optimizations
                         became.
  * ``ItemDataset``    — a list of per-item objects (variable-size images,
                         token lists, descriptor matrices).  Per-item ops map
                         over it; ``to_tensor`` stacks when shapes agree.

288 GB of HBM3E per GPU means shards stay resident: "cache" is simply *keep
the tensor*, and re-featurization (Spark lineage recompute) is an optimizer
decision, not a failure-recovery mechanism.
"""

from __future__ import annotations

from typing import Any, Callable, Iterable, List, Optional, Sequence

import torch

from .runtime import Runtime, get_runtime

__all__ = ["Dataset", "TensorDataset", "ItemDataset", "from_rows", "from_items"]


class Dataset:
    """A distributed dataset: this rank's shard plus the runtime handle."""

    def __init__(self, runtime: Optional[Runtime] = None):
        self.runtime = runtime or get_runtime()

    # -- interface -------------------------------------------------------
    def local_count(self) -> int:
        raise NotImplementedError

    def global_count(self) -> int:
        n = torch.tensor([self.local_count()], dtype=torch.int64)
        self.runtime.all_reduce_(n)
        return int(n.item())

    def map(self, fn: Callable[[Any], Any]) -> "Dataset":
        raise NotImplementedError

    def items(self) -> Iterable[Any]:
        raise NotImplementedError

    def first(self) -> Any:
        for x in self.items():
            return x
        raise IndexError("empty local shard")

    def cache(self) -> "Dataset":
        return self  # resident by construction; see workflow.cache.Cacher

    def sample_local(self, k: int, seed: int = 0) -> "Dataset":
        raise NotImplementedError


class TensorDataset(Dataset):
    """Row-sharded dense dataset: local shard is one ``[n_local, ...]`` tensor."""

    def __init__(self, data: torch.Tensor, runtime: Optional[Runtime] = None):
        super().__init__(runtime)
        self.data = data

    def local_count(self) -> int:
        return int(self.data.shape[0])

    def items(self):
        return iter(self.data)

    def map(self, fn) -> "ItemDataset":
        return ItemDataset([fn(x) for x in self.data], self.runtime)

    def map_batch(self, fn: Callable[[torch.Tensor], torch.Tensor]) -> "TensorDataset":
        return TensorDataset(fn(self.data), self.runtime)

    def sample_local(self, k: int, seed: int = 0) -> "TensorDataset":
        n = self.local_count()
        if k >= n:
            return self
        g = torch.Generator().manual_seed(seed + 7919 * self.runtime.rank)
        idx = torch.randperm(n, generator=g)[:k].to(self.data.device)
        return TensorDataset(self.data[idx], self.runtime)

    def collect(self) -> torch.Tensor:
        """All-gather the full matrix onto every rank (Spark ``collect``)."""
        return self.runtime.all_gather_rows(self.data)

    def to(self, device) -> "TensorDataset":
        return TensorDataset(self.data.to(device), self.runtime)


class ItemDataset(Dataset):
    """Per-item sharded dataset (images, token sequences, descriptor sets)."""

    def __init__(self, items: Sequence[Any], runtime: Optional[Runtime] = None):
        super().__init__(runtime)
        self._items = list(items)

    def local_count(self) -> int:
        return len(self._items)

    def items(self):
        return iter(self._items)

    def map(self, fn) -> "ItemDataset":
        return ItemDataset([fn(x) for x in self._items], self.runtime)

    def sample_local(self, k: int, seed: int = 0) -> "ItemDataset":
        import random
        if k >= len(self._items):
            return self
        rng = random.Random(seed + 7919 * self.runtime.rank)
        return ItemDataset(rng.sample(self._items, k), self.runtime)

    def to_tensor(self) -> TensorDataset:
        return TensorDataset(torch.stack(self._items), self.runtime)

    def collect(self) -> list:
        if not self.runtime.is_distributed:
            return list(self._items)
        import torch.distributed as dist
        gathered: List[Any] = [None] * self.runtime.world_size
        dist.all_gather_object(gathered, self._items)
        return [x for part in gathered for x in part]


def from_rows(data: torch.Tensor, runtime: Optional[Runtime] = None,
              already_sharded: bool = True) -> TensorDataset:
    """Wrap a local row block as this rank's shard.  When
    ``already_sharded`` is False, slice this rank's balanced share out of a
    replicated full tensor."""
    rt = runtime or get_runtime()
    if not already_sharded and rt.world_size > 1:
        lo, hi = rt.shard_bounds(int(data.shape[0]))
        data = data[lo:hi]
    return TensorDataset(data.to(rt.device), rt)


def from_items(items: Sequence[Any], runtime: Optional[Runtime] = None,
               already_sharded: bool = True) -> ItemDataset:
    rt = runtime or get_runtime()
    if not already_sharded and rt.world_size > 1:
        lo, hi = rt.shard_bounds(len(items))
        items = items[lo:hi]
    return ItemDataset(items, rt)
