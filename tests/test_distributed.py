"""Multi-process distributed semantics on CPU/gloo (ref: LocalSparkContext
local[2..4] test pattern).  Each test asserts the 2-rank fit
equals the single-process fit on the concatenated data — the invariant that
makes the RCCL path correct by construction."""

import pytest
import torch

from tests.dist_util import run_distributed

torch.manual_seed(0)
N, D, K = 64, 12, 3
X_ALL = torch.randn(N, D, dtype=torch.float64)
W_TRUE = torch.randn(D, K, dtype=torch.float64)
Y_ALL = X_ALL @ W_TRUE + 0.1 * torch.randn(N, K, dtype=torch.float64)
LAB_ALL = torch.randint(0, K, (N,))


def _shard(rank, world, t):
    base = N // world
    lo = rank * base
    hi = N if rank == world - 1 else lo + base
    return t[lo:hi]


def _linear_fit(rank, world):
    import keystone_amd as ka
    from keystone_amd.ops.learning import LinearMapEstimator
    ds = ka.from_rows(_shard(rank, world, X_ALL))
    ys = ka.from_rows(_shard(rank, world, Y_ALL))
    m = LinearMapEstimator(lam=0.5).fit(ds, ys)
    return m.W


def test_linear_map_distributed_matches_local():
    import keystone_amd as ka
    from keystone_amd.ops.learning import LinearMapEstimator
    ref = LinearMapEstimator(lam=0.5).fit(
        ka.from_rows(X_ALL), ka.from_rows(Y_ALL)).W
    results = run_distributed(_linear_fit, 2)
    for W in results:
        assert torch.allclose(W, ref, atol=1e-9)
    # ranks agree bitwise (replicated solve on all-reduced Gram)
    assert torch.equal(results[0], results[1])


def _block_fit(rank, world):
    import keystone_amd as ka
    from keystone_amd.ops.learning import BlockLeastSquaresEstimator
    ds = ka.from_rows(_shard(rank, world, X_ALL))
    ys = ka.from_rows(_shard(rank, world, Y_ALL))
    m = BlockLeastSquaresEstimator(block_size=5, num_iter=30, lam=0.5).fit(ds, ys)
    return torch.cat(m.blocks, dim=0), m.intercept


def test_block_ls_distributed_matches_local():
    import keystone_amd as ka
    from keystone_amd.ops.learning import BlockLeastSquaresEstimator
    ref = BlockLeastSquaresEstimator(block_size=5, num_iter=30, lam=0.5).fit(
        ka.from_rows(X_ALL), ka.from_rows(Y_ALL))
    ref_W = torch.cat(ref.blocks, dim=0)
    results = run_distributed(_block_fit, 2)
    for W, b in results:
        assert torch.allclose(W, ref_W, atol=1e-8)
        assert torch.allclose(b, ref.intercept, atol=1e-8)


def _weighted_fit(rank, world):
    import keystone_amd as ka
    from keystone_amd.ops.learning import BlockWeightedLeastSquaresEstimator
    Y_ind = torch.full((N, K), -1.0, dtype=torch.float64)
    Y_ind[torch.arange(N), LAB_ALL] = 1.0
    ds = ka.from_rows(_shard(rank, world, X_ALL))
    ys = ka.from_rows(_shard(rank, world, Y_ind))
    m = BlockWeightedLeastSquaresEstimator(
        block_size=6, num_iter=20, lam=1e-2, mixture_weight=0.5).fit(ds, ys)
    return torch.cat(m.blocks, dim=0)


def test_block_weighted_distributed_matches_local():
    import keystone_amd as ka
    from keystone_amd.ops.learning import BlockWeightedLeastSquaresEstimator
    Y_ind = torch.full((N, K), -1.0, dtype=torch.float64)
    Y_ind[torch.arange(N), LAB_ALL] = 1.0
    ref = BlockWeightedLeastSquaresEstimator(
        block_size=6, num_iter=20, lam=1e-2, mixture_weight=0.5).fit(
        ka.from_rows(X_ALL), ka.from_rows(Y_ind))
    ref_W = torch.cat(ref.blocks, dim=0)
    results = run_distributed(_weighted_fit, 2)
    for W in results:
        assert torch.allclose(W, ref_W, atol=1e-6)


def _scaler_fit(rank, world):
    import keystone_amd as ka
    from keystone_amd.ops.stats import StandardScaler
    ds = ka.from_rows(_shard(rank, world, X_ALL))
    m = StandardScaler().fit(ds)
    return m.mean, m.std


def test_standard_scaler_distributed():
    results = run_distributed(_scaler_fit, 2)
    mean_ref = X_ALL.mean(0)
    std_ref = X_ALL.std(0)
    for mean, std in results:
        assert torch.allclose(mean, mean_ref, atol=1e-10)
        assert torch.allclose(std, std_ref, atol=1e-10)


def _pca_fit(rank, world):
    import keystone_amd as ka
    from keystone_amd.ops.learning import DistributedPCAEstimator
    ds = ka.from_rows(_shard(rank, world, X_ALL))
    return DistributedPCAEstimator(4).fit(ds).components


def test_tsqr_pca_distributed_matches_local():
    import keystone_amd as ka
    from keystone_amd.ops.learning import PCAEstimator
    ref = PCAEstimator(4).fit(ka.from_rows(X_ALL)).components.to(torch.float64)
    results = run_distributed(_pca_fit, 2)
    for V in results:
        dots = (V.to(torch.float64).T @ ref).abs().diagonal()
        assert torch.allclose(dots, torch.ones(4, dtype=torch.float64), atol=1e-6)
    assert torch.equal(results[0], results[1])


def _lbfgs_fit(rank, world):
    import keystone_amd as ka
    from keystone_amd.ops.learning import DenseLBFGSwithL2, LeastSquaresGradient
    ds = ka.from_rows(_shard(rank, world, X_ALL).float())
    ys = ka.from_rows(_shard(rank, world, Y_ALL).float())
    return DenseLBFGSwithL2(LeastSquaresGradient(), 30, 1e-3).fit(ds, ys).W


def test_lbfgs_distributed_matches_local():
    import keystone_amd as ka
    from keystone_amd.ops.learning import DenseLBFGSwithL2, LeastSquaresGradient
    ref = DenseLBFGSwithL2(LeastSquaresGradient(), 30, 1e-3).fit(
        ka.from_rows(X_ALL.float()), ka.from_rows(Y_ALL.float())).W
    results = run_distributed(_lbfgs_fit, 2)
    for W in results:
        assert torch.allclose(W, ref, atol=1e-4)
    assert torch.equal(results[0], results[1])


def _eval_run(rank, world):
    import keystone_amd as ka
    from keystone_amd.evaluation import MulticlassClassifierEvaluator
    pred = _shard(rank, world, LAB_ALL)
    act = _shard(rank, world, LAB_ALL.roll(1))
    return MulticlassClassifierEvaluator(K).evaluate(
        ka.from_rows(pred), ka.from_rows(act)).confusion


def test_evaluator_distributed_confusion():
    from keystone_amd.evaluation import MulticlassClassifierEvaluator
    import keystone_amd as ka
    ref = MulticlassClassifierEvaluator(K).evaluate(
        ka.from_rows(LAB_ALL), ka.from_rows(LAB_ALL.roll(1))).confusion
    results = run_distributed(_eval_run, 2)
    for conf in results:
        assert torch.equal(conf, ref)


def _kmeans_fit(rank, world):
    import keystone_amd as ka
    from keystone_amd.ops.learning import KMeansPlusPlusEstimator
    ds = ka.from_rows(_shard(rank, world, X_ALL.float()))
    return KMeansPlusPlusEstimator(4, max_iterations=10, seed=2).fit(ds).centers


def test_kmeans_distributed_replicated():
    results = run_distributed(_kmeans_fit, 2)
    assert torch.equal(results[0], results[1])  # bitwise-replicated model


def _gmm_fit(rank, world):
    import keystone_amd as ka
    from keystone_amd.ops.learning.gmm import GaussianMixtureModelEstimator
    ds = ka.from_rows(_shard(rank, world, X_ALL.float()))
    m = GaussianMixtureModelEstimator(3, max_iterations=8, seed=1).fit(ds)
    return m.means, m.variances, m.weights


def test_gmm_distributed_replicated():
    results = run_distributed(_gmm_fit, 2)
    for a, b in zip(results[0], results[1]):
        assert torch.equal(a, b)


def _logistic_fit(rank, world):
    import keystone_amd as ka
    from keystone_amd.ops.learning import LogisticRegressionEstimator
    y = (X_ALL.float() @ torch.ones(D)) > 0
    ds = ka.from_rows(_shard(rank, world, X_ALL.float()))
    ys = ka.from_rows(_shard(rank, world, y.long()))
    return LogisticRegressionEstimator(2, 20, 1e-3).fit(ds, ys).W


def test_logistic_distributed_matches_local():
    import keystone_amd as ka
    from keystone_amd.ops.learning import LogisticRegressionEstimator
    y = (X_ALL.float() @ torch.ones(D)) > 0
    ref = LogisticRegressionEstimator(2, 20, 1e-3).fit(
        ka.from_rows(X_ALL.float()), ka.from_rows(y.long())).W
    results = run_distributed(_logistic_fit, 2)
    for W in results:
        assert torch.allclose(W, ref, atol=1e-4)


def _krr_fit_predict(rank, world):
    import keystone_amd as ka
    from keystone_amd.ops.learning import (GaussianKernelGenerator,
                                           KernelRidgeRegression)
    ds = ka.from_rows(_shard(rank, world, X_ALL.float()))
    ys = ka.from_rows(_shard(rank, world, Y_ALL.float()))
    m = KernelRidgeRegression(GaussianKernelGenerator(0.2), 1e-4, 16, 3).fit(ds, ys)
    out = m.apply_dataset(ds)
    return out.runtime.all_gather_rows(out.data)


def test_krr_distributed_matches_local():
    import keystone_amd as ka
    from keystone_amd.ops.learning import (GaussianKernelGenerator,
                                           KernelRidgeRegression)
    m = KernelRidgeRegression(GaussianKernelGenerator(0.2), 1e-4, 16, 3).fit(
        ka.from_rows(X_ALL.float()), ka.from_rows(Y_ALL.float()))
    ref = m.apply_batch(X_ALL.float())
    results = run_distributed(_krr_fit_predict, 2)
    for pred in results:
        assert torch.allclose(pred, ref, atol=1e-3)


def _async_reduce(rank, world):
    import torch as t
    from keystone_amd.parallel.runtime import get_runtime
    rt = get_runtime()
    xs = [t.full((4, 3), float(rank + i)) for i in range(3)]
    handles = [rt.all_reduce_async_(x) for x in xs]
    for h in handles:
        if h is not None:
            h.wait()
    return xs


def test_all_reduce_async_matches_sum():
    results = run_distributed(_async_reduce, 2)
    for i in range(3):
        expect = torch.full((4, 3), float(0 + i) + float(1 + i))
        assert torch.equal(results[0][i], expect)
        assert torch.equal(results[1][i], expect)


def _nlp_fit(rank, world, docs_all):
    import keystone_amd as ka
    from keystone_amd.ops.nlp import (CommonSparseFeatures, NGramsFeaturizer,
                                      StupidBackoffEstimator, TermFrequency,
                                      Tokenizer)
    from keystone_amd.parallel.dataset import ItemDataset
    from keystone_amd.parallel.runtime import get_runtime
    rt = get_runtime()
    base, rem = divmod(len(docs_all), world)
    lo = rank * base + min(rank, rem)
    hi = lo + base + (1 if rank < rem else 0)
    docs = ItemDataset(docs_all[lo:hi], rt)
    toks = Tokenizer().apply_dataset(docs)
    tf = TermFrequency().apply_dataset(
        NGramsFeaturizer(1, 2).apply_dataset(toks))
    vec = CommonSparseFeatures(30)._fit(tf)
    lm = StupidBackoffEstimator(max_order=2)._fit(toks)
    score = lm.apply(["the", "cat"])
    return sorted(vec.index.items(), key=lambda kv: kv[1]), float(score)


def test_nlp_featurization_distributed_matches_local():
    """CommonSparseFeatures top-K (reduceByKey analogue) and the backoff
    LM's merged counts at world 2 must equal the single-process fit."""
    docs = ["the cat sat on the mat", "the dog ate the cat food",
            "a cat and a dog", "the mat was red", "dogs eat food",
            "red cats sit"] * 3
    ref_idx, ref_score = _nlp_fit(0, 1, docs)
    results = run_distributed(_nlp_fit, 2, args=(docs,))
    for idx, score in results:
        assert idx == ref_idx
        assert abs(score - ref_score) < 1e-9


def _nb_fit(rank, world):
    import keystone_amd as ka
    from keystone_amd.ops.learning import NaiveBayesEstimator
    g = torch.Generator().manual_seed(21)
    X = (torch.rand(60, 12, generator=g) < 0.3).float() * \
        torch.randint(1, 5, (60, 12), generator=g)
    y = torch.randint(0, 3, (60,), generator=g)
    m = NaiveBayesEstimator(3, lam=1.0)._fit(
        ka.from_rows(_shard(rank, world, X)),
        ka.from_rows(_shard(rank, world, y)))
    return m.log_prior, m.log_theta


def test_naive_bayes_distributed_matches_local():
    lp_ref, lt_ref = _nb_fit(0, 1)
    for lp, lt in run_distributed(_nb_fit, 2):
        assert torch.allclose(lp, lp_ref, atol=1e-6)
        assert torch.allclose(lt, lt_ref, atol=1e-6)


def _ckpt_mixed_resume(rank, world, tmpdir):
    import keystone_amd as ka
    from keystone_amd.ops.learning import BlockLeastSquaresEstimator
    g = torch.Generator().manual_seed(8)
    X = torch.randn(40, 12, generator=g, dtype=torch.float64)
    Y = torch.randn(40, 2, generator=g, dtype=torch.float64)
    ds = ka.from_rows(_shard(rank, world, X))
    ys = ka.from_rows(_shard(rank, world, Y))
    est = BlockLeastSquaresEstimator(4, 6, 0.1, checkpoint_dir=tmpdir)
    m = est.fit(ds, ys)
    return torch.cat(m.blocks, dim=0)


def test_block_ls_refuses_mixed_epoch_checkpoints(tmp_path):
    """A crash between per-rank checkpoint saves leaves ranks at
    different epochs; resuming must fall back to a fresh fit (identical
    result), never silently blend mixed-epoch residuals."""
    import keystone_amd as ka
    from keystone_amd.ops.learning import BlockLeastSquaresEstimator
    g = torch.Generator().manual_seed(8)
    X = torch.randn(40, 12, generator=g, dtype=torch.float64)
    Y = torch.randn(40, 2, generator=g, dtype=torch.float64)
    ref = BlockLeastSquaresEstimator(4, 6, 0.1).fit(
        ka.from_rows(X), ka.from_rows(Y))
    ref_W = torch.cat(ref.blocks, dim=0)

    ck = tmp_path / "ck"
    ck.mkdir()
    # plant a stale epoch-3 checkpoint for rank 0 ONLY (rank 1 has none)
    stale = {"epoch": 3,
             "Ws": [torch.zeros(4, 2, dtype=torch.float64)] * 3,
             "R": torch.zeros(20, 2, dtype=torch.float64)}
    torch.save(stale, ck / "bls_rank0.pt")
    results = run_distributed(_ckpt_mixed_resume, 2, args=(str(ck),))
    for W in results:
        assert torch.allclose(W, ref_W, atol=1e-8)


def _ckpt_two_phase(rank, world, tmpdir):
    import keystone_amd as ka
    from keystone_amd.ops.learning import BlockLeastSquaresEstimator
    g = torch.Generator().manual_seed(8)
    X = torch.randn(40, 12, generator=g, dtype=torch.float64)
    Y = torch.randn(40, 2, generator=g, dtype=torch.float64)
    ds = ka.from_rows(_shard(rank, world, X))
    ys = ka.from_rows(_shard(rank, world, Y))
    # phase 1: run 3 epochs with checkpointing (all ranks save epoch 3)
    BlockLeastSquaresEstimator(4, 3, 0.1, checkpoint_dir=tmpdir).fit(ds, ys)
    # phase 2: resume to 6 epochs
    m = BlockLeastSquaresEstimator(4, 6, 0.1, checkpoint_dir=tmpdir).fit(ds, ys)
    return torch.cat(m.blocks, dim=0)


def test_block_ls_distributed_checkpoint_resume(tmp_path):
    """Clean resume at world 2 (every rank saved the same epoch) must
    equal the uninterrupted 6-epoch fit exactly."""
    import keystone_amd as ka
    from keystone_amd.ops.learning import BlockLeastSquaresEstimator
    g = torch.Generator().manual_seed(8)
    X = torch.randn(40, 12, generator=g, dtype=torch.float64)
    Y = torch.randn(40, 2, generator=g, dtype=torch.float64)
    ref = BlockLeastSquaresEstimator(4, 6, 0.1).fit(
        ka.from_rows(X), ka.from_rows(Y))
    ref_W = torch.cat(ref.blocks, dim=0)
    ck = tmp_path / "ck2"
    ck.mkdir()
    for W in run_distributed(_ckpt_two_phase, 2, args=(str(ck),)):
        assert torch.allclose(W, ref_W, atol=1e-10)


def test_gmm_distributed_replicated_world4():
    """World 4 with uneven shards: the vocab fit must stay bitwise
    replicated (the 8-GPU GMM path at a non-power world size)."""
    results = run_distributed(_gmm_fit, 4)
    for r in results[1:]:
        for a, b in zip(results[0], r):
            assert torch.equal(a, b)


def test_krr_distributed_world4_matches_local():
    import keystone_amd as ka
    from keystone_amd.ops.learning import (GaussianKernelGenerator,
                                           KernelRidgeRegression)
    m = KernelRidgeRegression(GaussianKernelGenerator(0.2), 1e-4, 16, 3).fit(
        ka.from_rows(X_ALL.float()), ka.from_rows(Y_ALL.float()))
    ref = m.apply_batch(X_ALL.float())
    for pred in run_distributed(_krr_fit_predict, 4):
        assert torch.allclose(pred, ref, atol=1e-3)


def _wbls_ckpt(rank, world, tmpdir):
    import keystone_amd as ka
    from keystone_amd.ops.learning import BlockWeightedLeastSquaresEstimator
    g = torch.Generator().manual_seed(15)
    X = torch.randn(48, 10, generator=g, dtype=torch.float64)
    y = torch.randint(0, 4, (48,), generator=g)
    Y = torch.full((48, 4), -1.0, dtype=torch.float64)
    Y[torch.arange(48), y] = 1.0
    m = BlockWeightedLeastSquaresEstimator(
        5, 8, 1e-2, 0.5, checkpoint_dir=tmpdir).fit(
        ka.from_rows(_shard(rank, world, X)),
        ka.from_rows(_shard(rank, world, Y)))
    return torch.cat(m.blocks, dim=0)


def test_weighted_refuses_mixed_epoch_checkpoints(tmp_path):
    """Weighted-solver flavor of the mixed-epoch guard: a stale rank-0
    checkpoint must be ignored (fresh fit), matching the no-checkpoint
    reference bitwise across ranks."""
    import keystone_amd as ka
    from keystone_amd.ops.learning import BlockWeightedLeastSquaresEstimator
    g = torch.Generator().manual_seed(15)
    X = torch.randn(48, 10, generator=g, dtype=torch.float64)
    y = torch.randint(0, 4, (48,), generator=g)
    Y = torch.full((48, 4), -1.0, dtype=torch.float64)
    Y[torch.arange(48), y] = 1.0
    ref = BlockWeightedLeastSquaresEstimator(5, 8, 1e-2, 0.5).fit(
        ka.from_rows(X), ka.from_rows(Y))
    ref_W = torch.cat(ref.blocks, dim=0)
    ck = tmp_path / "wck"
    ck.mkdir()
    stale = {"epoch": 5, "Ws": [torch.zeros(5, 4, dtype=torch.float64)] * 2,
             "R": torch.zeros(24, 4, dtype=torch.float64), "b_int": None}
    torch.save(stale, ck / "bwls_rank0.pt")
    for W in run_distributed(_wbls_ckpt, 2, args=(str(ck),)):
        assert torch.allclose(W, ref_W, atol=1e-10)
