"""ZCA whitening and linear discriminant analysis.

ref: nodes.learning.ZCAWhitenerEstimator / ZCAWhitener
(``W = V (Lam + eps I)^{-1/2} V^T`` from the eigendecomposition of the data
covariance; used by the CIFAR random-patch pipeline) and
LinearDiscriminantAnalysis (Fisher LDA projection).
"""

from __future__ import annotations


import torch

from ...parallel.dataset import TensorDataset
from ...workflow.pipeline import BatchTransformer, Estimator, LabelEstimator
from .pca import PCATransformer, _sign_fix

__all__ = ["ZCAWhitener", "ZCAWhitenerEstimator", "LinearDiscriminantAnalysis"]


class ZCAWhitener(BatchTransformer):
    def __init__(self, whitener: torch.Tensor, mean: torch.Tensor):
        self.whitener = whitener    # [d, d]
        self.mean = mean            # [d]

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        self.whitener = self.whitener.to(device=X.device, dtype=X.dtype)
        self.mean = self.mean.to(device=X.device, dtype=X.dtype)
        return (X - self.mean) @ self.whitener

    @classmethod
    def from_saved(cls, config, state):
        return cls(state["whitener"], state["mean"])


class ZCAWhitenerEstimator(Estimator):
    def __init__(self, eps: float = 0.1):
        self.eps = eps

    def _fit(self, data: TensorDataset) -> ZCAWhitener:
        X = data.data
        rt = data.runtime
        n = torch.tensor([float(X.shape[0])], dtype=torch.float64, device=X.device)
        s = X.sum(0, dtype=torch.float64)
        cov = X.to(torch.float64).T @ X.to(torch.float64)
        rt.all_reduce_many_([n, s, cov])
        n_tot = n.item()
        mu = s / n_tot
        cov = cov / n_tot - torch.outer(mu, mu)
        evals, evecs = torch.linalg.eigh(cov)
        W = evecs @ torch.diag((evals.clamp_min(0) + self.eps).rsqrt()) @ evecs.T
        return ZCAWhitener(W.to(X.dtype), mu.to(X.dtype))


class LinearDiscriminantAnalysis(LabelEstimator):
    """Fisher LDA: project onto the top eigenvectors of Sw^{-1} Sb
    (ref: LinearDiscriminantAnalysis(numDimensions))."""

    def __init__(self, num_dimensions: int, eps: float = 1e-6):
        self.num_dimensions = num_dimensions
        self.eps = eps

    def _fit(self, data: TensorDataset, labels: TensorDataset) -> PCATransformer:
        X = data.data.to(torch.float64)
        y = labels.data
        if y.dim() > 1:
            y = y.argmax(dim=1)
        y = y.long()
        rt = data.runtime
        K = int(y.max().item()) + 1
        Kt = torch.tensor([K], dtype=torch.int64)
        rt.all_reduce_(Kt, op="max")
        K = int(Kt.item())
        d = X.shape[1]

        n = torch.tensor([float(X.shape[0])], dtype=torch.float64, device=X.device)
        class_sum = torch.zeros(K, d, dtype=torch.float64, device=X.device)
        class_sum.index_add_(0, y, X)
        class_n = torch.bincount(y, minlength=K).to(torch.float64)
        xx = X.T @ X
        total_sum = X.sum(0)
        rt.all_reduce_many_([n, class_sum, class_n, xx, total_sum])

        n_tot = n.item()
        gmean = total_sum / n_tot
        cmeans = class_sum / class_n.clamp_min(1.0).unsqueeze(1)
        # between-class scatter
        diff = cmeans - gmean
        Sb = (diff.T * class_n) @ diff
        # within-class scatter = total scatter - between
        St = xx - n_tot * torch.outer(gmean, gmean)
        Sw = St - Sb + self.eps * torch.eye(d, dtype=torch.float64, device=X.device)
        evals, evecs = torch.linalg.eig(torch.linalg.solve(Sw, Sb))
        order = evals.real.argsort(descending=True)
        V = _sign_fix(evecs.real[:, order[: self.num_dimensions]])
        return PCATransformer(V.to(data.data.dtype),
                              gmean.to(data.data.dtype))
