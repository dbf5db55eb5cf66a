"""Kernel-matrix block abstractions.

ref (SURVEY.md §2.2 [M]): nodes.learning.KernelMatrix / BlockKernelMatrix —
lazy views over the (n x n) kernel matrix, materialized one row-block at a
time (the full matrix never exists).  Used by KernelRidgeRegression; kept
as a public abstraction for parity and for custom kernel solvers.
"""

from __future__ import annotations

from typing import Callable, Tuple

import torch

from ...parallel.dataset import TensorDataset

__all__ = ["KernelMatrix", "BlockKernelMatrix"]


class KernelMatrix:
    """Lazy distributed kernel matrix K(X, X): ``block(lo, hi)`` gathers
    the block's rows (all-gather over ranks) and returns this rank's
    column slice K[lo:hi, local]."""

    def __init__(self, kernel_fn: Callable, data: TensorDataset):
        self.kernel_fn = kernel_fn
        self.data = data
        rt = data.runtime
        counts = rt.all_gather(torch.tensor([data.local_count()],
                                            dtype=torch.int64))
        self.counts = [int(c.item()) for c in counts]
        self.offsets = [sum(self.counts[:r]) for r in range(len(self.counts))]
        self.n_total = sum(self.counts)

    def local_range(self) -> Tuple[int, int]:
        r = self.data.runtime.rank
        return self.offsets[r], self.offsets[r] + self.counts[r]

    def block_rows(self, lo: int, hi: int) -> torch.Tensor:
        """Gather global rows [lo, hi) of X (xGMI all-gather)."""
        rt = self.data.runtime
        my_lo, my_hi = self.local_range()
        sl = slice(max(lo, my_lo) - my_lo, max(min(hi, my_hi), my_lo) - my_lo)
        part = self.data.data[sl]
        if not rt.is_distributed:
            return part
        return rt.all_gather_rows(part)

    def block(self, lo: int, hi: int) -> torch.Tensor:
        """K[lo:hi, local_cols] — the block kernel slab this rank owns."""
        Xb = self.block_rows(lo, hi)
        return self.kernel_fn(Xb, self.data.data)


class BlockKernelMatrix(KernelMatrix):
    """Iterator flavour: yields (lo, hi, K_block_local) over row blocks."""

    def __init__(self, kernel_fn, data: TensorDataset, block_size: int = 2048):
        super().__init__(kernel_fn, data)
        self.block_size = block_size

    def __iter__(self):
        lo = 0
        while lo < self.n_total:
            hi = min(lo + self.block_size, self.n_total)
            yield lo, hi, self.block(lo, hi)
            lo = hi
