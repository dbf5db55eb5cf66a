"""Vector plumbing nodes.

ref: nodes.util.VectorCombiner (concat gather-branch
outputs), VectorSplitter (split d-dim features into blocks — the
reference's feature-block "model parallel" axis feeding the block solvers),
Densify / Sparsify, FloatToDouble (generalized to Cast).
"""

from __future__ import annotations

from typing import List

import torch

from ...parallel.dataset import Dataset, TensorDataset
from ...workflow.pipeline import BatchTransformer, Transformer, TupleDataset


class VectorCombiner(Transformer):
    """Concatenate a tuple of vectors (gather output) along the last dim."""

    def apply(self, xs) -> torch.Tensor:
        return torch.cat(list(xs), dim=-1)

    def apply_dataset(self, ds: Dataset) -> Dataset:
        if isinstance(ds, TupleDataset) and all(
                isinstance(p, TensorDataset) for p in ds.parts):
            return TensorDataset(
                torch.cat([p.data for p in ds.parts], dim=-1), ds.runtime)
        return ds.map(self.apply)


class VectorSplitter(Transformer):
    """Split a d-dim feature vector into ``ceil(d/block)`` blocks
    (ref: VectorSplitter — feeds BlockLinearMapper / block solvers)."""

    def __init__(self, block_size: int):
        self.block_size = block_size

    def apply(self, x: torch.Tensor) -> List[torch.Tensor]:
        return list(torch.split(x, self.block_size, dim=-1))

    def apply_dataset(self, ds: Dataset) -> Dataset:
        if isinstance(ds, TensorDataset):
            parts = torch.split(ds.data, self.block_size, dim=-1)
            return TupleDataset([TensorDataset(p, ds.runtime) for p in parts])
        return ds.map(self.apply)


class Densify(BatchTransformer):
    """sparse -> dense (ref: Densify)."""

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        return X.to_dense() if X.is_sparse or X.layout == torch.sparse_csr else X

    def apply(self, x):
        return x.to_dense() if x.is_sparse or x.layout == torch.sparse_csr else x


class Sparsify(BatchTransformer):
    """dense -> sparse CSR (ref: Sparsify)."""

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        return X.to_sparse_csr()

    def apply(self, x):
        return x.to_sparse()


class Cast(BatchTransformer):
    """dtype cast (generalizes ref: FloatToDouble)."""

    def __init__(self, dtype: str = "float64"):
        self.dtype = dtype

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        return X.to(getattr(torch, self.dtype))
