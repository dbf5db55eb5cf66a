"""Logistic regression and naive Bayes (the text-pipeline classifiers).

ref: nodes.learning.LogisticRegressionEstimator (wrapped
Spark MLlib LogisticRegressionWithLBFGS; used by Amazon/Newsgroups) and
NaiveBayesEstimator (wrapped MLlib NaiveBayes).  Rebuilt on our own
distributed L-BFGS / all-reduced count aggregation — no MLlib.
"""

from __future__ import annotations

import torch

from ...parallel.dataset import TensorDataset
from ...workflow.pipeline import BatchTransformer, LabelEstimator
from .lbfgs import DenseLBFGSwithL2, LogisticGradient, SparseLBFGSwithL2
from .linear import LinearMapper

__all__ = ["LogisticRegressionEstimator", "LogisticRegressionModel",
           "NaiveBayesEstimator", "NaiveBayesModel"]


class LogisticRegressionModel(BatchTransformer):
    """Predicts the argmax class (KeystoneML's wrapped MLlib model predicted
    labels; use ``scores=True`` at fit time for raw scores instead)."""

    def __init__(self, W: torch.Tensor, b: torch.Tensor | None = None,
                 return_scores: bool = False):
        self.W = W
        self.b = b
        self.return_scores = return_scores

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        self.W = self.W.to(device=X.device, dtype=X.dtype if X.dtype.is_floating_point else self.W.dtype)
        S = X @ self.W
        if self.b is not None:
            self.b = self.b.to(device=X.device, dtype=self.W.dtype)
            S = S + self.b
        return S if self.return_scores else S.argmax(dim=-1)

    @classmethod
    def from_saved(cls, config, state):
        return cls(state["W"], state.get("b"),
                   config.get("return_scores", False))


class LogisticRegressionEstimator(LabelEstimator):
    def __init__(self, num_classes: int, num_iters: int = 50,
                 reg_param: float = 0.0, fit_intercept: bool = True,
                 scores: bool = False, sparse: str = "auto"):
        self.num_classes = num_classes
        self.num_iters = num_iters
        self.reg_param = reg_param
        self.fit_intercept = fit_intercept
        self.scores = scores
        self.sparse = sparse

    def _fit(self, data: TensorDataset, labels: TensorDataset):
        X = data.data
        Y = labels.data
        if Y.dim() == 1:
            Y = torch.nn.functional.one_hot(Y.long(),
                                            self.num_classes).to(torch.float32)
        is_sparse = X.layout == torch.sparse_csr or X.is_sparse \
            if self.sparse == "auto" else self.sparse == "true"
        solver_cls = SparseLBFGSwithL2 if is_sparse else DenseLBFGSwithL2
        solver = solver_cls(LogisticGradient(), self.num_iters,
                            self.reg_param, fit_intercept=self.fit_intercept)
        lm: LinearMapper = solver._fit(data, TensorDataset(Y, labels.runtime))
        return LogisticRegressionModel(lm.W, lm.b, self.scores)


class NaiveBayesModel(BatchTransformer):
    """Outputs per-class log-posterior scores (follow with MaxClassifier)."""

    def __init__(self, log_prior: torch.Tensor, log_theta: torch.Tensor):
        self.log_prior = log_prior   # [K]
        self.log_theta = log_theta   # [d, K]

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        self.log_theta = self.log_theta.to(device=X.device)
        self.log_prior = self.log_prior.to(device=X.device)
        lt, lp = self.log_theta, self.log_prior
        if X.layout == torch.sparse_csr or X.is_sparse:
            return torch.sparse.mm(X, lt.to(X.dtype)) + lp.to(X.dtype)
        return X @ lt.to(X.dtype) + lp.to(X.dtype)

    @classmethod
    def from_saved(cls, config, state):
        return cls(state["log_prior"], state["log_theta"])


class NaiveBayesEstimator(LabelEstimator):
    """Multinomial naive Bayes (ref: NaiveBayesEstimator(lambda)).
    Per-class feature-count sums -> one all-reduce -> replicated log-model."""

    def __init__(self, num_classes: int, lam: float = 1.0):
        self.num_classes = num_classes
        self.lam = lam

    def _fit(self, data: TensorDataset, labels: TensorDataset) -> NaiveBayesModel:
        X = data.data
        y = labels.data
        if y.dim() > 1:
            y = y.argmax(dim=1)
        y = y.long()
        rt = data.runtime
        K = self.num_classes
        if X.layout == torch.sparse_csr or X.is_sparse:
            X = X.to_dense()
        d = X.shape[1]
        counts = torch.zeros(K, d, dtype=torch.float64, device=X.device)
        counts.index_add_(0, y, X.to(torch.float64))
        class_n = torch.bincount(y, minlength=K).to(torch.float64)
        rt.all_reduce_many_([counts, class_n])
        log_theta = (counts + self.lam).log() - \
            (counts.sum(1, keepdim=True) + self.lam * d).log()
        log_prior = (class_n.clamp_min(1e-12) / class_n.sum()).log()
        return NaiveBayesModel(log_prior.to(torch.float32),
                               log_theta.T.contiguous().to(torch.float32))
