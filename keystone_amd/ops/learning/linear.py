"""Linear models and exact least-squares solvers.

ref: nodes.learning.LinearMapper, LinearMapEstimator,
LocalLeastSquaresEstimator.

Comm pattern: Spark's per-partition GEMM + ``treeReduce(+)`` of the Gram
``A^T A`` and ``A^T B`` becomes per-rank GEMM + ONE fused RCCL all-reduce
over xGMI; the "driver Cholesky" is a *replicated local solve* on every
rank (deterministic, zero extra communication — cheaper than broadcast on
a 7-link point-to-point mesh).
"""

from __future__ import annotations

from typing import Optional

import torch

from ...parallel.dataset import TensorDataset
from ...workflow.pipeline import BatchTransformer, LabelEstimator


def _solve_dtype(dt: torch.dtype) -> torch.dtype:
    """Factorization dtype: fp64 inputs stay fp64 (oracle parity); fp32
    inputs solve in fp32 — rocSOLVER/rocBLAS fp32 runs ~10x the fp64 rate
    on MI355X and the BCD solvers re-derive the residual exactly each
    pass, so per-block fp32 solve error is self-correcting."""
    return torch.float64 if dt == torch.float64 else torch.float32


def _solve_spd(gram: torch.Tensor, rhs: torch.Tensor,
               lam: float) -> torch.Tensor:
    """Solve (gram + lam I) W = rhs, return input dtype.  Replicated on
    every rank — inputs are all-reduced, so outputs agree bitwise across
    ranks.  On fp32 Cholesky failure, retries with jitter then fp64."""
    dt = gram.dtype
    sdt = _solve_dtype(dt)
    eye = torch.eye(gram.shape[0], dtype=sdt, device=gram.device)
    G = gram.to(sdt)
    if lam:
        G = G + lam * eye
    try:
        L = torch.linalg.cholesky(G)
    except torch.linalg.LinAlgError:
        try:
            jitter = 1e-6 * float(G.diagonal().abs().mean())
            L = torch.linalg.cholesky(G + jitter * eye)
        except torch.linalg.LinAlgError:
            # the fp32 GRAM itself can be non-PSD when feature magnitudes
            # make its accumulation error exceed lam (observed: CIFAR
            # sum-pooled conv features, diag ~1e9 -> error ~1e2 >> lam);
            # casting to fp64 cannot repair that, so jitter scaled to the
            # diagonal is part of the fp64 fallback as well
            eye64 = eye.to(torch.float64)
            G64 = gram.to(torch.float64) + lam * eye64
            # jitter LADDER: the fp32 Gram's accumulation error scales
            # with its diagonal (observed -1e2..-1e4 eigenvalues at diag
            # ~3e8 on CIFAR sum-pooled conv features) — one fixed step is
            # not enough when the user's lam is tiny relative to the
            # feature scale
            j64 = 1e-8 * float(G64.diagonal().abs().mean())
            last = None
            for _ in range(6):
                try:
                    L64 = torch.linalg.cholesky(G64 + j64 * eye64)
                    break
                except torch.linalg.LinAlgError as e:
                    last = e
                    j64 *= 32.0
            else:
                raise last
            return torch.cholesky_solve(rhs.to(torch.float64), L64).to(dt)
    W = torch.cholesky_solve(rhs.to(sdt), L)
    return W.to(dt)


class LinearMapper(BatchTransformer):
    """Apply ``x -> x W + b`` (ref: LinearMapper). Weights stay resident on
    the compute device; the per-shard apply is ONE library GEMM
    (hipBLASLt/rocBLAS via torch.matmul on ROCm)."""

    def __init__(self, W: torch.Tensor, b: Optional[torch.Tensor] = None):
        self.W = W
        self.b = b

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        self.W = self.W.to(device=X.device, dtype=X.dtype)
        out = X @ self.W
        if self.b is not None:
            self.b = self.b.to(device=X.device, dtype=X.dtype)
            out = out + self.b
        return out

    @classmethod
    def from_saved(cls, config, state):
        return cls(state["W"], state.get("b"))


class LinearMapEstimator(LabelEstimator):
    """Exact ridge least squares via normal equations (ref:
    LinearMapEstimator): Gram + cross-moment per rank, fused all-reduce,
    replicated fp64 Cholesky."""

    def __init__(self, lam: float = 0.0, fit_intercept: bool = False):
        self.lam = lam
        self.fit_intercept = fit_intercept

    def _fit(self, data: TensorDataset, labels: TensorDataset) -> LinearMapper:
        X, Y = data.data, labels.data
        if Y.dim() == 1:
            Y = Y.unsqueeze(1)
        Y = Y.to(X.dtype)
        rt = data.runtime

        x_mean = y_mean = None
        if self.fit_intercept:
            n = torch.tensor([float(X.shape[0])], dtype=torch.float64,
                             device=X.device)
            sx = X.sum(0, dtype=torch.float64)
            sy = Y.sum(0, dtype=torch.float64)
            rt.all_reduce_many_([n, sx, sy])
            x_mean = (sx / n).to(X.dtype)
            y_mean = (sy / n).to(X.dtype)
            X = X - x_mean
            Y = Y - y_mean

        gram = X.T @ X
        xty = X.T @ Y
        rt.all_reduce_many_([gram, xty])
        W = _solve_spd(gram, xty, self.lam)
        b = (y_mean - x_mean @ W) if self.fit_intercept else None
        return LinearMapper(W, b)


class LocalLeastSquaresEstimator(LabelEstimator):
    """Collect-and-solve for small problems (ref: LocalLeastSquaresEstimator):
    all-gather the full (X, Y) and run one replicated lstsq."""

    def __init__(self, lam: float = 0.0):
        self.lam = lam

    def _fit(self, data: TensorDataset, labels: TensorDataset) -> LinearMapper:
        X = data.collect().to(torch.float64)
        Y = labels.collect().to(torch.float64)
        if Y.dim() == 1:
            Y = Y.unsqueeze(1)
        if self.lam:
            d = X.shape[1]
            X = torch.cat([X, (self.lam ** 0.5) *
                           torch.eye(d, dtype=X.dtype, device=X.device)])
            Y = torch.cat([Y, torch.zeros(d, Y.shape[1], dtype=Y.dtype,
                                          device=Y.device)])
        W = torch.linalg.lstsq(X, Y).solution
        return LinearMapper(W.to(data.data.dtype))
