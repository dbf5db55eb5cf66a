"""Distributed batch L-BFGS and SGD solvers.

ref: nodes.learning.DenseLBFGSwithL2, SparseLBFGSwithL2,
Gradient (LeastSquaresGradient / LogisticGradient), MiniBatchSGDwithL2.

Spark's per-iteration ``treeAggregate`` of the gradient (size d x k)
becomes one fused RCCL all-reduce of (loss, gradient); the L-BFGS two-loop
recursion and line search run replicated on every rank on the all-reduced
quantities, so the model stays bitwise-identical across ranks with zero
extra communication.
"""

from __future__ import annotations

from typing import List, Tuple

import torch

from ...parallel.dataset import TensorDataset
from ...workflow.pipeline import LabelEstimator
from .linear import LinearMapper

__all__ = ["LeastSquaresGradient", "LogisticGradient", "DenseLBFGSwithL2",
           "SparseLBFGSwithL2", "MiniBatchSGDwithL2"]


class Gradient:
    """loss_local, grad_local for the local shard; all-reduced by the solver."""

    def compute(self, W: torch.Tensor, X: torch.Tensor,
                Y: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        raise NotImplementedError


class LeastSquaresGradient(Gradient):
    """0.5 ||XW - Y||^2 (ref: LeastSquaresDenseGradient)."""

    def compute(self, W, X, Y):
        E = X @ W - Y
        loss = 0.5 * (E * E).sum()
        grad = _xt_mm(X, E)
        return loss, grad


class LogisticGradient(Gradient):
    """Multinomial softmax cross-entropy (ref: LogisticGradient / MLlib
    LogisticRegressionWithLBFGS semantics)."""

    def compute(self, W, X, Y):
        S = X @ W                       # [n, k] scores
        logZ = torch.logsumexp(S, dim=1, keepdim=True)
        P = torch.exp(S - logZ)
        loss = (logZ.squeeze(1) - (S * Y).sum(1)).sum()
        grad = _xt_mm(X, P - Y)
        return loss, grad


def _xt_mm(X: torch.Tensor, D: torch.Tensor) -> torch.Tensor:
    """X^T @ D for dense or sparse-CSR X."""
    if X.layout == torch.sparse_csr or X.is_sparse:
        # CSR has no cheap transpose-matmul; go through COO transpose
        return torch.sparse.mm(X.to_sparse_coo().t(), D)
    return X.T @ D


class _LBFGSBase(LabelEstimator):
    def __init__(self, gradient: Gradient, num_iterations: int = 50,
                 reg_param: float = 0.0, history: int = 10,
                 tol: float = 1e-7, fit_intercept: bool = False):
        self.gradient = gradient
        self.num_iterations = num_iterations
        self.reg_param = reg_param
        self.history = history
        self.tol = tol
        self.fit_intercept = fit_intercept

    # ---------------------------------------------------------------
    def _prepare(self, X):
        return X

    def _fit(self, data: TensorDataset, labels: TensorDataset) -> LinearMapper:
        X = self._prepare(data.data)
        Y = labels.data
        if Y.dim() == 1:
            Y = torch.nn.functional.one_hot(Y.long()).to(torch.float32)
        dtype = torch.float32 if (X.dtype not in
                                  (torch.float32, torch.float64)) else X.dtype
        Y = Y.to(dtype)
        rt = data.runtime
        d = X.shape[1] + (1 if self.fit_intercept else 0)
        if self.fit_intercept:
            ones = torch.ones(X.shape[0], 1, dtype=dtype, device=Y.device)
            X = torch.cat([X.to_dense() if X.layout == torch.sparse_csr else X,
                           ones], dim=1)
        k = Y.shape[1]
        dev = Y.device
        W = torch.zeros(d, k, dtype=dtype, device=dev)

        n = torch.tensor([float(Y.shape[0])], dtype=torch.float64, device=dev)
        rt.all_reduce_(n)
        n_tot = n.item()

        def value_grad(Wc):
            loss, grad = self.gradient.compute(Wc, X, Y)
            buf = torch.cat([loss.reshape(1).to(torch.float64),
                             grad.reshape(-1).to(torch.float64)])
            rt.all_reduce_(buf)
            loss_g = buf[0].item() / n_tot
            grad_g = (buf[1:].reshape(d, k) / n_tot).to(dtype)
            if self.reg_param:
                loss_g += 0.5 * self.reg_param * float((Wc.to(torch.float64) ** 2).sum())
                grad_g = grad_g + self.reg_param * Wc
            return loss_g, grad_g

        s_hist: List[torch.Tensor] = []
        y_hist: List[torch.Tensor] = []
        rho_hist: List[float] = []
        loss, grad = value_grad(W)
        for _it in range(self.num_iterations):
            # two-loop recursion (replicated; identical on all ranks)
            q = grad.clone().reshape(-1)
            alphas = []
            for s, y, rho in zip(reversed(s_hist), reversed(y_hist),
                                 reversed(rho_hist)):
                a = rho * float(s @ q)
                q -= a * y
                alphas.append(a)
            if y_hist:
                y_last = y_hist[-1]
                gamma = float(s_hist[-1] @ y_last) / max(float(y_last @ y_last), 1e-30)
                q *= gamma
            for (s, y, rho), a in zip(zip(s_hist, y_hist, rho_hist),
                                      reversed(alphas)):
                b = rho * float(y @ q)
                q += (a - b) * s
            direction = -q.reshape(d, k)

            # backtracking Armijo line search on the global objective
            g_dot_d = float((grad * direction).sum())
            if g_dot_d > 0:  # not a descent direction; reset
                direction = -grad
                g_dot_d = -float((grad * grad).sum())
                s_hist, y_hist, rho_hist = [], [], []
            step = 1.0
            for _ls in range(20):
                W_new = W + step * direction
                loss_new, grad_new = value_grad(W_new)
                if loss_new <= loss + 1e-4 * step * g_dot_d:
                    break
                step *= 0.5
            s_vec = (W_new - W).reshape(-1)
            y_vec = (grad_new - grad).reshape(-1)
            sy = float(s_vec @ y_vec)
            if sy > 1e-12:
                s_hist.append(s_vec)
                y_hist.append(y_vec)
                rho_hist.append(1.0 / sy)
                if len(s_hist) > self.history:
                    s_hist.pop(0), y_hist.pop(0), rho_hist.pop(0)
            if abs(loss - loss_new) <= self.tol * max(1.0, abs(loss)):
                W, loss, grad = W_new, loss_new, grad_new
                break
            W, loss, grad = W_new, loss_new, grad_new

        if self.fit_intercept:
            return LinearMapper(W[:-1], W[-1])
        return LinearMapper(W)


class DenseLBFGSwithL2(_LBFGSBase):
    """ref: DenseLBFGSwithL2."""


class SparseLBFGSwithL2(_LBFGSBase):
    """ref: SparseLBFGSwithL2 — X as torch sparse CSR; gradient GEMMs go
    through sparse kernels (rocSPARSE via torch on ROCm)."""

    def _prepare(self, X):
        if X.layout != torch.sparse_csr and not X.is_sparse:
            return X.to_sparse_csr()
        return X


class MiniBatchSGDwithL2(LabelEstimator):
    """ref (SURVEY.md §2.2 [L]): MiniBatchSGDwithL2 — synchronous SGD with
    per-step all-reduced gradients."""

    def __init__(self, gradient: Gradient, num_iterations: int = 100,
                 step_size: float = 0.1, batch_fraction: float = 0.1,
                 reg_param: float = 0.0, seed: int = 0):
        self.gradient = gradient
        self.num_iterations = num_iterations
        self.step_size = step_size
        self.batch_fraction = batch_fraction
        self.reg_param = reg_param
        self.seed = seed

    def _fit(self, data: TensorDataset, labels: TensorDataset) -> LinearMapper:
        X, Y = data.data, labels.data
        if Y.dim() == 1:
            Y = torch.nn.functional.one_hot(Y.long()).to(torch.float32)
        Y = Y.to(X.dtype)
        rt = data.runtime
        n_local, d = X.shape
        k = Y.shape[1]
        W = torch.zeros(d, k, dtype=X.dtype, device=X.device)
        g = torch.Generator().manual_seed(self.seed + 7919 * rt.rank)
        bs = max(1, int(n_local * self.batch_fraction))
        for it in range(self.num_iterations):
            idx = torch.randperm(n_local, generator=g)[:bs].to(X.device)
            _loss, grad = self.gradient.compute(W, X[idx], Y[idx])
            grad = grad.to(torch.float64)
            rt.all_reduce_(grad)
            grad = grad.to(X.dtype) / (bs * rt.world_size)
            if self.reg_param:
                grad = grad + self.reg_param * W
            W -= (self.step_size / (1 + it) ** 0.5) * grad
        return LinearMapper(W)
