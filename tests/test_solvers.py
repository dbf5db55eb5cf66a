"""Solver numerics vs local oracles (ref test strategy: SURVEY.md §4 —
every solver checked against a dense local computation, Stats.aboutEq
style; BlockLeastSquares converges to the exact normal-equation solution)."""

import math

import pytest
import torch

import keystone_amd as ka
from keystone_amd.ops.learning import (BlockLeastSquaresEstimator,
                                       BlockWeightedLeastSquaresEstimator,
                                       DenseLBFGSwithL2,
                                       GaussianKernelGenerator,
                                       KernelRidgeRegression,
                                       KMeansPlusPlusEstimator,
                                       LeastSquaresEstimator,
                                       LeastSquaresGradient,
                                       LinearMapEstimator,
                                       LocalLeastSquaresEstimator,
                                       LogisticGradient,
                                       LogisticRegressionEstimator,
                                       NaiveBayesEstimator,
                                       SparseLBFGSwithL2)
from keystone_amd.ops.learning.gmm import GaussianMixtureModelEstimator
from keystone_amd.ops.learning.pca import (ApproximatePCAEstimator,
                                           DistributedPCAEstimator,
                                           PCAEstimator)
from keystone_amd.ops.learning.whitening import (LinearDiscriminantAnalysis,
                                                 ZCAWhitenerEstimator)

torch.manual_seed(0)


def ridge_oracle(X, Y, lam):
    X64, Y64 = X.to(torch.float64), Y.to(torch.float64)
    d = X.shape[1]
    G = X64.T @ X64 + lam * torch.eye(d, dtype=torch.float64)
    return torch.linalg.solve(G, X64.T @ Y64)


def test_linear_map_estimator_exact():
    X = torch.randn(64, 12, dtype=torch.float64)
    W_true = torch.randn(12, 3, dtype=torch.float64)
    Y = X @ W_true
    model = LinearMapEstimator(lam=0.1).fit(ka.from_rows(X), ka.from_rows(Y))
    assert torch.allclose(model.W, ridge_oracle(X, Y, 0.1), atol=1e-8)


def test_linear_map_estimator_intercept():
    X = torch.randn(128, 6, dtype=torch.float64)
    W_true = torch.randn(6, 2, dtype=torch.float64)
    Y = X @ W_true + torch.tensor([1.0, -2.0], dtype=torch.float64)
    m = LinearMapEstimator(lam=0.0, fit_intercept=True).fit(
        ka.from_rows(X), ka.from_rows(Y))
    pred = m.apply_batch(X)
    assert torch.allclose(pred, Y, atol=1e-6)


def test_local_least_squares():
    X = torch.randn(40, 8, dtype=torch.float64)
    Y = torch.randn(40, 2, dtype=torch.float64)
    m = LocalLeastSquaresEstimator(lam=0.5).fit(ka.from_rows(X), ka.from_rows(Y))
    assert torch.allclose(m.W, ridge_oracle(X, Y, 0.5), atol=1e-7)


def test_block_least_squares_converges_to_exact():
    X = torch.randn(200, 24, dtype=torch.float64)
    W_true = torch.randn(24, 4, dtype=torch.float64)
    Y = X @ W_true + 0.01 * torch.randn(200, 4, dtype=torch.float64)
    lam = 0.5
    m = BlockLeastSquaresEstimator(block_size=8, num_iter=60, lam=lam,
                                   fit_intercept=False).fit(
        ka.from_rows(X), ka.from_rows(Y))
    W_bcd = torch.cat(m.blocks, dim=0)
    assert torch.allclose(W_bcd, ridge_oracle(X, Y, lam), atol=1e-5)


def test_block_least_squares_intercept():
    X = torch.randn(300, 16, dtype=torch.float64)
    W_true = torch.randn(16, 2, dtype=torch.float64)
    b_true = torch.tensor([3.0, -1.0], dtype=torch.float64)
    Y = X @ W_true + b_true
    m = BlockLeastSquaresEstimator(block_size=8, num_iter=50, lam=1e-8).fit(
        ka.from_rows(X), ka.from_rows(Y))
    pred = m.apply_batch(X)
    assert torch.allclose(pred, Y, atol=1e-4)


def test_block_weighted_reduces_to_plain_when_balanced():
    # alpha -> 0 makes every weight (1-alpha)/n = uniform: same solution as
    # unweighted BCD with rescaled lambda
    n, d, K = 120, 12, 3
    X = torch.randn(n, d, dtype=torch.float64)
    y = torch.randint(0, K, (n,))
    Y = torch.full((n, K), -1.0, dtype=torch.float64)
    Y[torch.arange(n), y] = 1.0
    lam = 1e-3
    m_w = BlockWeightedLeastSquaresEstimator(
        block_size=6, num_iter=40, lam=lam, mixture_weight=1e-12,
        fit_intercept=False).fit(
        ka.from_rows(X), ka.from_rows(Y))
    W_w = torch.cat(m_w.blocks, dim=0)
    # weighted normal equations at alpha=0: (G/n + lam I) W = X^T Y / n
    W_ref = ridge_oracle(X, Y, lam * n)
    assert torch.allclose(W_w, W_ref, atol=1e-5)


def test_block_weighted_oracle_per_class():
    # direct check against the per-column weighted normal equations
    n, d, K = 90, 10, 3
    X = torch.randn(n, d, dtype=torch.float64)
    y = torch.cat([torch.zeros(60), torch.ones(20), 2 * torch.ones(10)]).long()
    Y = torch.full((n, K), -1.0, dtype=torch.float64)
    Y[torch.arange(n), y] = 1.0
    alpha, lam = 0.5, 1e-2
    m = BlockWeightedLeastSquaresEstimator(
        block_size=d, num_iter=30, lam=lam, mixture_weight=alpha,
        fit_intercept=False).fit(
        ka.from_rows(X), ka.from_rows(Y))
    W = torch.cat(m.blocks, dim=0)
    counts = torch.bincount(y, minlength=K).to(torch.float64)
    for c in range(K):
        w = torch.full((n,), (1 - alpha) / n, dtype=torch.float64)
        w[y == c] += alpha / counts[c]
        G = X.T @ (w.unsqueeze(1) * X) + lam * torch.eye(d, dtype=torch.float64)
        rhs = X.T @ (w * Y[:, c])
        ref = torch.linalg.solve(G, rhs)
        assert torch.allclose(W[:, c], ref, atol=1e-5), f"class {c}"


def test_lbfgs_least_squares():
    X = torch.randn(150, 10)
    W_true = torch.randn(10, 2)
    Y = X @ W_true
    m = DenseLBFGSwithL2(LeastSquaresGradient(), num_iterations=80,
                         reg_param=0.0).fit(ka.from_rows(X), ka.from_rows(Y))
    assert torch.allclose(m.W, W_true, atol=1e-2)


def test_logistic_regression_separable():
    n = 200
    X = torch.randn(n, 5)
    w = torch.randn(5)
    y = (X @ w > 0).long()
    model = LogisticRegressionEstimator(num_classes=2, num_iters=60,
                                        reg_param=1e-4).fit(
        ka.from_rows(X), ka.from_rows(y))
    pred = model.apply_batch(X)
    acc = (pred == y).float().mean().item()
    assert acc > 0.97


def test_sparse_lbfgs_matches_dense():
    X = (torch.rand(100, 20) < 0.3).float() * torch.randn(100, 20)
    Y = torch.randn(100, 2)
    dense = DenseLBFGSwithL2(LeastSquaresGradient(), 40, 1e-3).fit(
        ka.from_rows(X), ka.from_rows(Y))
    sparse = SparseLBFGSwithL2(LeastSquaresGradient(), 40, 1e-3).fit(
        ka.from_rows(X.to_sparse_csr()), ka.from_rows(Y))
    assert torch.allclose(dense.W, sparse.W, atol=1e-4)


def test_naive_bayes_oracle():
    X = torch.poisson(torch.rand(80, 12) * 3)
    y = torch.randint(0, 3, (80,))
    model = NaiveBayesEstimator(num_classes=3, lam=1.0).fit(
        ka.from_rows(X), ka.from_rows(y))
    # oracle
    counts = torch.zeros(3, 12, dtype=torch.float64)
    counts.index_add_(0, y, X.to(torch.float64))
    lt = (counts + 1).log() - (counts.sum(1, keepdim=True) + 12).log()
    scores = model.apply_batch(X)
    ref = X @ lt.T.float() + (torch.bincount(y, minlength=3).float() / 80).log()
    assert torch.allclose(scores, ref, atol=1e-4)


def test_kmeans_recovers_clusters():
    centers = torch.tensor([[5.0, 5.0], [-5.0, -5.0], [5.0, -5.0]])
    X = torch.cat([centers[i] + 0.2 * torch.randn(50, 2) for i in range(3)])
    model = KMeansPlusPlusEstimator(3, max_iterations=30, seed=1).fit(
        ka.from_rows(X))
    C = model.centers
    d = torch.cdist(C.double(), centers.double())
    assert float(d.min(dim=1).values.max()) < 0.3
    onehot = model.apply_batch(X)
    assert onehot.shape == (150, 3)
    assert torch.allclose(onehot.sum(1), torch.ones(150))


def test_gmm_recovers_mixture():
    torch.manual_seed(3)
    mu = torch.tensor([[4.0, 0.0], [-4.0, 0.0]])
    X = torch.cat([mu[0] + torch.randn(300, 2) * 0.5,
                   mu[1] + torch.randn(300, 2) * 1.0])
    gmm = GaussianMixtureModelEstimator(2, max_iterations=60, seed=0).fit(
        ka.from_rows(X))
    order = gmm.means[:, 0].argsort(descending=True)
    assert torch.allclose(gmm.means[order], mu, atol=0.2)
    assert torch.allclose(gmm.weights.sum(), torch.tensor(1.0), atol=1e-5)
    gamma = gmm.apply_batch(X)
    assert torch.allclose(gamma.sum(1), torch.ones(600), atol=1e-5)


def test_pca_matches_svd_oracle():
    X = torch.randn(100, 8) @ torch.diag(torch.tensor([5., 4., 3., 2., 1., .5, .2, .1]))
    pca = PCAEstimator(3).fit(ka.from_rows(X))
    _u, _s, vh = torch.linalg.svd(X.to(torch.float64), full_matrices=False)
    Vo = vh.T[:, :3]
    # compare up to sign
    dots = (pca.components.to(torch.float64).T @ Vo).abs().diagonal()
    assert torch.allclose(dots, torch.ones(3, dtype=torch.float64), atol=1e-5)


def test_distributed_pca_matches_local_single_rank():
    X = torch.randn(60, 6)
    p1 = PCAEstimator(4).fit(ka.from_rows(X))
    p2 = DistributedPCAEstimator(4).fit(ka.from_rows(X))
    dots = (p1.components.T.to(torch.float64)
            @ p2.components.to(torch.float64)).abs().diagonal()
    assert torch.allclose(dots, torch.ones(4, dtype=torch.float64), atol=1e-4)


def test_approximate_pca_close():
    X = torch.randn(200, 30) @ torch.randn(30, 30)
    exact = PCAEstimator(5).fit(ka.from_rows(X))
    approx = ApproximatePCAEstimator(5, oversample=10, power_iters=2).fit(
        ka.from_rows(X))
    dots = (exact.components.T.to(torch.float64)
            @ approx.components.to(torch.float64)).abs().diagonal()
    assert float(dots.min()) > 0.95


def test_zca_whitening_identity_cov():
    X = torch.randn(500, 6) @ torch.randn(6, 6)
    zca = ZCAWhitenerEstimator(eps=1e-6).fit(ka.from_rows(X))
    Z = zca.apply_batch(X)
    cov = (Z.T @ Z / Z.shape[0]).to(torch.float64)
    assert torch.allclose(cov, torch.eye(6, dtype=torch.float64), atol=0.05)


def test_lda_separates_classes():
    mu = torch.tensor([[3.0, 0.0, 0.0], [-3.0, 0.0, 0.0]])
    X = torch.cat([mu[0] + torch.randn(100, 3), mu[1] + torch.randn(100, 3)])
    y = torch.cat([torch.zeros(100), torch.ones(100)]).long()
    lda = LinearDiscriminantAnalysis(1).fit(ka.from_rows(X), ka.from_rows(y))
    proj = lda.apply_batch(X).squeeze(1)
    m0, m1 = proj[:100].mean(), proj[100:].mean()
    s = proj[:100].std() + proj[100:].std()
    assert abs(float(m0 - m1)) / float(s) > 1.5


def test_krr_fits_nonlinear_function():
    torch.manual_seed(1)
    X = torch.rand(200, 1) * 6 - 3
    Y = torch.sin(X)
    kern = GaussianKernelGenerator(gamma=1.0)
    model = KernelRidgeRegression(kern, lam=1e-6, block_size=64,
                                  num_epochs=4).fit(ka.from_rows(X), ka.from_rows(Y))
    pred = model.apply_batch(X)
    err = (pred - Y).abs().max().item()
    assert err < 0.05


def test_least_squares_dispatcher():
    X = torch.randn(50, 8)
    Y = torch.randn(50, 2)
    ds, ys = ka.from_rows(X), ka.from_rows(Y)
    est = LeastSquaresEstimator(lam=0.1)
    chosen = est.optimize(ds, ys)
    assert isinstance(chosen, LocalLeastSquaresEstimator)
    est2 = LeastSquaresEstimator(lam=0.1, dense_solve_max=4)
    chosen2 = est2.optimize(ds, ys)
    assert isinstance(chosen2, BlockLeastSquaresEstimator)
    chosen3 = LeastSquaresEstimator().optimize(
        ka.from_rows(X.to_sparse_csr()), ys)
    assert isinstance(chosen3, SparseLBFGSwithL2)


def test_block_weighted_woodbury_path_matches_direct():
    # n_c << d_b forces the Woodbury rank-update path; compare against the
    # per-column weighted normal equations oracle
    torch.manual_seed(7)
    n, d, K = 40, 32, 8          # n_c ~ 5 rows << d_b = 32
    X = torch.randn(n, d, dtype=torch.float64)
    y = torch.arange(n) % K
    Y = torch.full((n, K), -1.0, dtype=torch.float64)
    Y[torch.arange(n), y] = 1.0
    alpha, lam = 0.4, 1e-3
    m = BlockWeightedLeastSquaresEstimator(
        block_size=d, num_iter=25, lam=lam, mixture_weight=alpha,
        woodbury_threshold=0.5, fit_intercept=False).fit(ka.from_rows(X), ka.from_rows(Y))
    W = torch.cat(m.blocks, dim=0)
    counts = torch.bincount(y, minlength=K).to(torch.float64)
    for c in range(K):
        w = torch.full((n,), (1 - alpha) / n, dtype=torch.float64)
        w[y == c] += alpha / counts[c]
        G = X.T @ (w.unsqueeze(1) * X) + lam * torch.eye(d, dtype=torch.float64)
        ref = torch.linalg.solve(G, X.T @ (w * Y[:, c]))
        assert torch.allclose(W[:, c], ref, atol=1e-6), f"class {c}"


def test_lazy_block_featurization_matches_materialized():
    """ref capability: BlockLinearMapper lazy per-block featurization —
    features are regenerated per block instead of materialized jointly."""
    from keystone_amd.ops.learning.block_ls import LazyBlocks
    from keystone_amd.ops.stats import CosineRandomFeatures
    torch.manual_seed(11)
    n, d, bs, nb = 80, 12, 16, 3
    X = torch.randn(n, d)
    Y = torch.randn(n, 2)
    ds = ka.from_rows(X)
    feats = [CosineRandomFeatures(d, bs, 0.5, seed=100 + i) for i in range(nb)]
    lazy = LazyBlocks(ds, feats)
    m_lazy = BlockLeastSquaresEstimator(bs, 8, 1e-3).fit(lazy, ka.from_rows(Y))
    # materialized reference
    Xf = torch.cat([f.apply_batch(X) for f in feats], dim=1)
    m_mat = BlockLeastSquaresEstimator(bs, 8, 1e-3).fit(
        ka.from_rows(Xf), ka.from_rows(Y))
    for a, b in zip(m_lazy.blocks, m_mat.blocks):
        assert torch.allclose(a, b, atol=1e-4)
    # lazy apply: takes RAW input, never the 48-dim features
    pred_lazy = m_lazy.apply_batch(X)
    pred_mat = m_mat.apply_batch(Xf)
    assert torch.allclose(pred_lazy, pred_mat, atol=1e-4)
    # dataset-level lazy apply + save/load round trip
    out = m_lazy.apply_dataset(ds).data
    assert torch.allclose(out, pred_mat, atol=1e-4)


def test_block_weighted_intercept_oracle():
    # unpenalized intercept: per column c the oracle solves the augmented
    # weighted normal equations with a constant column and zero lam on it
    torch.manual_seed(11)
    n, d, K = 80, 9, 3
    X = torch.randn(n, d, dtype=torch.float64) + 0.7   # nonzero means
    y = torch.cat([torch.zeros(50), torch.ones(20), 2 * torch.ones(10)]).long()
    Y = torch.full((n, K), -1.0, dtype=torch.float64)
    Y[torch.arange(n), y] = 1.0
    alpha, lam = 0.5, 1e-2
    m = BlockWeightedLeastSquaresEstimator(
        block_size=5, num_iter=80, lam=lam, mixture_weight=alpha,
        fit_intercept=True).fit(ka.from_rows(X), ka.from_rows(Y))
    W = torch.cat(m.blocks, dim=0)
    counts = torch.bincount(y, minlength=K).to(torch.float64)
    for c in range(K):
        w = torch.full((n,), (1 - alpha) / n, dtype=torch.float64)
        w[y == c] += alpha / counts[c]
        Xa = torch.cat([X, torch.ones(n, 1, dtype=torch.float64)], dim=1)
        reg = torch.eye(d + 1, dtype=torch.float64) * lam
        reg[d, d] = 0.0
        G = Xa.T @ (w.unsqueeze(1) * Xa) + reg
        ref = torch.linalg.solve(G, Xa.T @ (w * Y[:, c]))
        assert torch.allclose(W[:, c], ref[:d], atol=1e-5), f"class {c} W"
        assert torch.allclose(m.intercept[c], ref[d], atol=1e-5), f"class {c} b"


def test_block_weighted_lazy_blocks_and_checkpoint(tmp_path):
    # LazyBlocks input (blocks regenerated per epoch) + epoch checkpoints
    from keystone_amd.ops.learning import LazyBlocks
    from keystone_amd.workflow.pipeline import Transformer

    class TakeCols(Transformer):
        def __init__(self, lo, hi):
            self.lo, self.hi = lo, hi

        def apply(self, x):
            return x[..., self.lo:self.hi]

        def apply_batch(self, X):
            return X[:, self.lo:self.hi]

    torch.manual_seed(3)
    n, d, K = 60, 12, 3
    X = torch.randn(n, d, dtype=torch.float64)
    y = torch.arange(n) % K
    Y = torch.full((n, K), -1.0, dtype=torch.float64)
    Y[torch.arange(n), y] = 1.0
    raw = ka.from_rows(X)
    lazy = LazyBlocks(raw, [TakeCols(0, 6), TakeCols(6, 12)])
    kw = dict(block_size=6, num_iter=25, lam=1e-2, mixture_weight=0.5,
              fit_intercept=True)
    m_ref = BlockWeightedLeastSquaresEstimator(**kw).fit(
        ka.from_rows(X), ka.from_rows(Y))
    m_lazy = BlockWeightedLeastSquaresEstimator(
        **kw, checkpoint_dir=str(tmp_path)).fit(lazy, ka.from_rows(Y))
    assert torch.allclose(torch.cat(m_lazy.blocks), torch.cat(m_ref.blocks),
                          atol=1e-10)
    assert torch.allclose(m_lazy.intercept, m_ref.intercept, atol=1e-10)
    # lazy mapper keeps featurizers: applies to RAW input
    pred_lazy = m_lazy.apply(X)
    pred_ref = m_ref.apply(X)
    assert torch.allclose(pred_lazy, pred_ref, atol=1e-10)
    # resume: checkpoint recorded the last epoch; a rerun with more epochs
    # continues (smoke: it loads and completes without error, same answer
    # as a fresh fit with the same total epochs)
    import os
    assert os.path.exists(tmp_path / "bwls_rank0.pt")
    m_more = BlockWeightedLeastSquaresEstimator(
        block_size=6, num_iter=40, lam=1e-2, mixture_weight=0.5,
        fit_intercept=True, checkpoint_dir=str(tmp_path)).fit(
        lazy, ka.from_rows(Y))
    m_fresh = BlockWeightedLeastSquaresEstimator(
        block_size=6, num_iter=40, lam=1e-2, mixture_weight=0.5,
        fit_intercept=True).fit(lazy, ka.from_rows(Y))
    assert torch.allclose(torch.cat(m_more.blocks), torch.cat(m_fresh.blocks),
                          atol=1e-9)


def test_block_weighted_rejects_multilabel():
    n, d, K = 20, 6, 3
    X = torch.randn(n, d, dtype=torch.float64)
    Y = torch.full((n, K), -1.0, dtype=torch.float64)
    Y[:, 0] = 1.0
    Y[:, 1] = 1.0  # two positives per row: VOC-style multi-label
    import pytest
    with pytest.raises(ValueError, match="single-label"):
        BlockWeightedLeastSquaresEstimator(block_size=3, num_iter=2).fit(
            ka.from_rows(X), ka.from_rows(Y))


def test_minibatch_sgd_converges():
    from keystone_amd.ops.learning import LeastSquaresGradient
    from keystone_amd.ops.learning.lbfgs import MiniBatchSGDwithL2
    torch.manual_seed(12)
    X = torch.randn(400, 8)
    W_true = torch.randn(8, 2)
    Y = X @ W_true
    m = MiniBatchSGDwithL2(LeastSquaresGradient(), num_iterations=1500,
                           step_size=0.1, batch_fraction=0.5,
                           reg_param=0.0).fit(ka.from_rows(X),
                                              ka.from_rows(Y))
    rel = (m.apply_batch(X) - Y).norm() / Y.norm()
    assert rel.item() < 0.05, rel.item()


def test_lbfgs_matches_ridge_oracle_with_l2():
    """Regularized fixed point: the objective is (1/n)·0.5||XW−Y||² +
    0.5·reg·||W||², so the minimizer solves (XᵀX/n + reg·I) W = XᵀY/n.
    Guards against the lam-scaling bug family (cf. the KRR block-update
    fix) that a reg=0 test cannot see."""
    g = torch.Generator().manual_seed(3)
    n, d, k, reg = 120, 8, 3, 0.3
    X = torch.randn(n, d, generator=g)
    Y = torch.randn(n, k, generator=g)
    m = DenseLBFGSwithL2(LeastSquaresGradient(), num_iterations=200,
                         reg_param=reg).fit(ka.from_rows(X), ka.from_rows(Y))
    Xd, Yd = X.to(torch.float64), Y.to(torch.float64)
    ref = torch.linalg.solve(
        Xd.T @ Xd / n + reg * torch.eye(d, dtype=torch.float64),
        Xd.T @ Yd / n)
    assert torch.allclose(m.W.to(torch.float64), ref, atol=1e-3), \
        (m.W.to(torch.float64) - ref).abs().max()


def test_logistic_matches_autograd_oracle():
    """Same objective ((1/n)·softmax-CE + 0.5·reg·||W||², bias column
    augmented and regularized like the solver does) optimized with torch
    autograd LBFGS in fp64 must land on the same W."""
    g = torch.Generator().manual_seed(11)
    n, d, K, reg = 150, 6, 3, 0.05
    X = torch.randn(n, d, generator=g)
    y = torch.randint(0, K, (n,), generator=g)
    model = LogisticRegressionEstimator(num_classes=K, num_iters=300,
                                        reg_param=reg).fit(
        ka.from_rows(X), ka.from_rows(y))
    W_ours = torch.cat([model.W, model.b.unsqueeze(0)]).to(torch.float64)

    Xa = torch.cat([X, torch.ones(n, 1)], dim=1).to(torch.float64)
    Yh = torch.nn.functional.one_hot(y, K).to(torch.float64)
    W = torch.zeros(d + 1, K, dtype=torch.float64, requires_grad=True)
    opt = torch.optim.LBFGS([W], max_iter=400, tolerance_grad=1e-12,
                            line_search_fn="strong_wolfe")

    def closure():
        opt.zero_grad()
        S = Xa @ W
        loss = (torch.logsumexp(S, dim=1) - (S * Yh).sum(1)).mean() \
            + 0.5 * reg * (W ** 2).sum()
        loss.backward()
        return loss

    opt.step(closure)
    ref = W.detach()
    # softmax W has a shift gauge only without regularization; with reg>0
    # the optimum is unique, compare directly
    assert torch.allclose(W_ours, ref, atol=2e-3), \
        (W_ours - ref).abs().max()


def test_gmm_em_fixed_point_oracle():
    """At convergence the fitted (mu, var, w) must satisfy the EM
    fixed-point equations with posteriors recomputed from scratch via the
    textbook diagonal-Gaussian Bayes formula in fp64 — validates both the
    E-step (log-prob/softmax) and M-step (weighted MLE) algebra, which
    the recovery test alone cannot."""
    torch.manual_seed(7)
    mu0 = torch.tensor([[3.0, -1.0, 0.0], [-2.0, 2.0, 1.0], [0.0, 0.0, -3.0]])
    X = torch.cat([mu0[i] + torch.randn(250, 3) * (0.5 + 0.3 * i)
                   for i in range(3)])
    gmm = GaussianMixtureModelEstimator(3, max_iterations=200, tol=1e-12,
                                        seed=1).fit(ka.from_rows(X))
    Xd = X.to(torch.float64)
    m = gmm.means.to(torch.float64)
    v = gmm.variances.to(torch.float64)
    w = gmm.weights.to(torch.float64)
    # hand E-step
    lp = (-0.5 * ((Xd.unsqueeze(1) - m) ** 2 / v).sum(2)
          - 0.5 * v.log().sum(1) - 0.5 * 3 * torch.log(
              torch.tensor(2 * torch.pi, dtype=torch.float64))
          + w.log())
    gam = torch.softmax(lp, dim=1)
    # model's own posteriors must match the textbook formula
    assert torch.allclose(gmm.apply_batch(X).to(torch.float64), gam,
                          atol=1e-5)
    # hand M-step from those posteriors reproduces the model (fixed point)
    Nk = gam.sum(0)
    mu_next = gam.T @ Xd / Nk.unsqueeze(1)
    var_next = gam.T @ (Xd ** 2) / Nk.unsqueeze(1) - mu_next ** 2
    assert torch.allclose(m, mu_next, atol=1e-3), (m - mu_next).abs().max()
    assert torch.allclose(v, var_next, atol=1e-3)
    assert torch.allclose(w, Nk / X.shape[0], atol=1e-4)
