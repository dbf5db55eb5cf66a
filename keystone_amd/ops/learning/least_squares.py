"""Cost-model least-squares dispatcher — the flagship Optimizable node.

ref: nodes.learning.LeastSquaresEstimator — an
``OptimizableLabelEstimator`` that chooses among physical solver
implementations (exact normal equations, block coordinate descent, L-BFGS,
collect-and-solve) from problem statistics (n, d, k, sparsity, cluster
size) — the node-level half of the ICDE'17 whole-pipeline optimizer.

The cost model is re-derived for one MI355X node: per-GPU HBM is 288 GB, a
dense fp32 Gram solve is cheap up to d ~ 16k (d^2 fp32 = 1 GB at 16k, and
rocSOLVER/torch Cholesky at that size is sub-second), the all-reduce cost
grows with d^2 (Gram) vs per-iteration d*k (LBFGS) — so:

  * sparse features              -> SparseLBFGSwithL2
  * d <= dense_solve_max (16384) -> LinearMapEstimator (one-shot Gram)
  * wider                        -> BlockLeastSquaresEstimator (4096 blocks)
"""

from __future__ import annotations


import torch

from ...parallel.dataset import Dataset, TensorDataset
from ...workflow.optimizer import Optimizable
from ...workflow.pipeline import LabelEstimator, TupleDataset
from .block_ls import BlockLeastSquaresEstimator
from .lbfgs import LeastSquaresGradient, SparseLBFGSwithL2
from .linear import LinearMapEstimator, LocalLeastSquaresEstimator

__all__ = ["LeastSquaresEstimator"]


class LeastSquaresEstimator(LabelEstimator, Optimizable):
    def __init__(self, lam: float = 0.0, num_iter: int = 10,
                 block_size: int = 4096, dense_solve_max: int = 16384,
                 local_solve_max_rows: int = 4096):
        self.lam = lam
        self.num_iter = num_iter
        self.block_size = block_size
        self.dense_solve_max = dense_solve_max
        self.local_solve_max_rows = local_solve_max_rows

    # node-level physical-operator selection (ref: Optimizable*.optimize)
    def optimize(self, data: Dataset, labels=None) -> LabelEstimator:
        if isinstance(data, TupleDataset):
            return BlockLeastSquaresEstimator(self.block_size, self.num_iter,
                                              self.lam)
        if not isinstance(data, TensorDataset):
            return LinearMapEstimator(self.lam)
        X = data.data
        sparse = X.layout == torch.sparse_csr or X.is_sparse
        d = X.shape[1]
        n_global = data.global_count()
        if sparse:
            return SparseLBFGSwithL2(LeastSquaresGradient(),
                                     num_iterations=self.num_iter * 5,
                                     reg_param=self.lam)
        if n_global <= self.local_solve_max_rows and d <= self.dense_solve_max:
            return LocalLeastSquaresEstimator(self.lam)
        if d <= self.dense_solve_max:
            return LinearMapEstimator(self.lam)
        return BlockLeastSquaresEstimator(self.block_size, self.num_iter,
                                          self.lam)

    def _fit(self, data, labels):
        return self.optimize(data, labels).fit(data, labels)
