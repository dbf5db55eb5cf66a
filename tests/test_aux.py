"""Aux subsystem tests: checkpoint/resume, determinism,
tracing, config, CLI plumbing."""

import json
import os

import pytest
import subprocess
import sys

import torch

import keystone_amd as ka
from keystone_amd.ops.learning import BlockLeastSquaresEstimator
from keystone_amd.utils.determinism import model_fingerprint
from keystone_amd.utils import tracing

torch.manual_seed(0)


def test_solver_checkpoint_resume(tmp_path):
    X = torch.randn(100, 12, dtype=torch.float64)
    Y = torch.randn(100, 2, dtype=torch.float64)
    ds, ys = ka.from_rows(X), ka.from_rows(Y)
    # full run: 6 epochs straight
    full = BlockLeastSquaresEstimator(4, 6, 0.1).fit(ds, ys)
    # interrupted run: 3 epochs with checkpointing, then resume to 6
    ck = str(tmp_path / "ck")
    BlockLeastSquaresEstimator(4, 3, 0.1, checkpoint_dir=ck).fit(ds, ys)
    assert os.path.exists(os.path.join(ck, "bls_rank0.pt"))
    resumed = BlockLeastSquaresEstimator(4, 6, 0.1, checkpoint_dir=ck).fit(ds, ys)
    for a, b in zip(full.blocks, resumed.blocks):
        assert torch.allclose(a, b, atol=1e-12)


def test_model_fingerprint_deterministic():
    X = torch.randn(50, 8)
    Y = torch.randn(50, 2)
    from keystone_amd.ops.learning import LinearMapEstimator
    m1 = LinearMapEstimator(0.1).fit(ka.from_rows(X), ka.from_rows(Y))
    m2 = LinearMapEstimator(0.1).fit(ka.from_rows(X), ka.from_rows(Y))
    assert model_fingerprint(m1) == model_fingerprint(m2)
    m3 = LinearMapEstimator(0.2).fit(ka.from_rows(X), ka.from_rows(Y))
    assert model_fingerprint(m1) != model_fingerprint(m3)


def test_refit_determinism_pipeline():
    """Same seed => bitwise-identical fitted pipeline (re-run check)."""
    from keystone_amd.ops.stats import CosineRandomFeatures
    from keystone_amd.ops.learning import BlockLeastSquaresEstimator
    X = torch.randn(60, 10)
    Y = torch.randn(60, 3)
    ds, ys = ka.from_rows(X), ka.from_rows(Y)

    def fit_once():
        pipe = (CosineRandomFeatures(10, 64, 0.5, seed=7)
                .and_then(BlockLeastSquaresEstimator(16, 3, 1e-3), ds, ys))
        return pipe.fit()

    f1, f2 = fit_once(), fit_once()
    fps1 = [model_fingerprint(t) for t in f1.transformers()]
    fps2 = [model_fingerprint(t) for t in f2.transformers()]
    assert sorted(fps1) == sorted(fps2)


def test_trace_jsonl_and_chrome(tmp_path):
    trace_file = str(tmp_path / "trace.jsonl")
    tracing.enable_trace(trace_file)
    try:
        from keystone_amd.ops.stats import NormalizeRows
        ds = ka.from_rows(torch.randn(5, 4))
        NormalizeRows().and_then(ka.Identity()).apply(ds).get()
    finally:
        tracing.disable_trace()
    lines = [json.loads(l) for l in open(trace_file)]
    assert any(l["kind"] == "op" for l in lines)
    chrome = str(tmp_path / "trace.json")
    tracing.export_chrome_trace(trace_file, chrome)
    data = json.load(open(chrome))
    assert data["traceEvents"]


def test_runtime_config_env(monkeypatch):
    from keystone_amd.utils import config
    monkeypatch.setenv("KEYSTONE_BLOCK_SIZE", "1234")
    monkeypatch.setenv("KEYSTONE_HBM_BUDGET_GB", "100.5")
    cfg = config.RuntimeConfig.from_env()
    assert cfg.block_size == 1234
    assert cfg.hbm_budget_gb == 100.5


def test_cli_mnist_smoke():
    out = subprocess.run(
        [sys.executable, "bin/run_pipeline.py", "mnist", "--n", "300",
         "--num-features", "256"],
        capture_output=True, text=True, timeout=300,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert out.returncode == 0, out.stderr[-2000:]
    assert "accuracy" in out.stdout


@pytest.mark.parametrize("cmd", [
    ["newsgroups", "--n-docs", "200", "--num-classes", "4",
     "--common-features", "500"],
    ["amazon", "--n-docs", "200"],
    ["timit", "--n", "1500", "--num-features", "512", "--epochs", "1"],
    ["imagenet-fv", "--images", "12", "--image-size", "64",
     "--num-classes", "4", "--gmm-k", "4"],
])
def test_cli_apps_smoke(cmd):
    """Every reference example app's CLI twin runs end-to-end on tiny
    synthetic data (mnist/cifar/voc covered elsewhere)."""
    out = subprocess.run(
        [sys.executable, "bin/run_pipeline.py"] + cmd,
        capture_output=True, text=True, timeout=600,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert out.returncode == 0, out.stderr[-2000:]
    assert "accuracy" in out.stdout or "mAP" in out.stdout


def test_state_cache_reuses_fit(tmp_path):
    from keystone_amd.workflow import state_cache
    from keystone_amd.ops.learning import LinearMapEstimator

    calls = []

    class CountingEst(LinearMapEstimator):
        def _fit(self, data, labels):
            calls.append(1)
            return super()._fit(data, labels)

    X = torch.randn(30, 5)
    Y = torch.randn(30, 2)
    ds, ys = ka.from_rows(X), ka.from_rows(Y)
    state_cache.enable_state_cache(str(tmp_path / "sc"))
    try:
        p1 = ka.Identity().and_then(CountingEst(0.1), ds, ys).fit()
        p2 = ka.Identity().and_then(CountingEst(0.1), ds, ys).fit()
        assert len(calls) == 1  # second fit loaded from the state cache
        a = p1.apply(ds).get().data
        b = p2.apply(ds).get().data
        assert torch.allclose(a, b, atol=1e-6)
    finally:
        state_cache.disable_state_cache()


def test_state_cache_does_not_alias_different_data(tmp_path):
    """identical estimators fit on DIFFERENT datasets
    must not return each other's cached models — the cache key folds in a
    fit-data fingerprint."""
    from keystone_amd.workflow import state_cache
    from keystone_amd.ops.learning import LinearMapEstimator
    torch.manual_seed(3)
    X1, Y1 = torch.randn(30, 5), torch.randn(30, 2)
    X2, Y2 = torch.randn(30, 5), torch.randn(30, 2)
    state_cache.enable_state_cache(str(tmp_path / "sc2"))
    try:
        p1 = ka.Identity().and_then(LinearMapEstimator(0.1),
                                    ka.from_rows(X1), ka.from_rows(Y1)).fit()
        p2 = ka.Identity().and_then(LinearMapEstimator(0.1),
                                    ka.from_rows(X2), ka.from_rows(Y2)).fit()
        a = p1.apply(ka.from_rows(X1)).get().data
        b = p2.apply(ka.from_rows(X1)).get().data
        assert not torch.allclose(a, b, atol=1e-6)  # distinct fits
    finally:
        state_cache.disable_state_cache()


def test_shuffler_preserves_rows():
    from keystone_amd.ops.util import Shuffler
    X = torch.arange(20, dtype=torch.float32).reshape(10, 2)
    out = Shuffler(seed=1).apply_dataset(ka.from_rows(X))
    assert out.data.shape == X.shape
    assert torch.equal(out.data.sum(0), X.sum(0))


def test_kernel_matrix_blocks():
    from keystone_amd.ops.learning import BlockKernelMatrix, GaussianKernelGenerator
    X = torch.randn(30, 4)
    km = BlockKernelMatrix(GaussianKernelGenerator(0.5), ka.from_rows(X), 8)
    full = GaussianKernelGenerator(0.5)(X, X)
    got = torch.cat([blk for _lo, _hi, blk in km], dim=0)
    assert torch.allclose(got, full, atol=1e-6)


def test_no_silent_fallback_contract(monkeypatch):
    """with the extension missing, GPU dispatch must raise
    unless KEYSTONE_ALLOW_EAGER=1 explicitly authorizes the eager path."""
    import keystone_amd.hip as hip
    monkeypatch.setattr(hip, "_lib", None)
    monkeypatch.setattr(hip, "_LIB_PATH", "/nonexistent/libkeystone_hip.so")
    monkeypatch.delenv("KEYSTONE_ALLOW_EAGER", raising=False)
    import pytest as _pytest
    with _pytest.raises(RuntimeError, match="refusing"):
        hip.gpu_ready()
    monkeypatch.setenv("KEYSTONE_ALLOW_EAGER", "1")
    assert hip.gpu_ready() is False


@pytest.mark.gpu
def test_gpu_op_raises_without_extension(monkeypatch):
    """On a real GPU: a dispatching op must fail loudly when the extension
    is hidden (not silently produce the eager result)."""
    import keystone_amd.hip as hip
    from keystone_amd.ops.stats import CosineRandomFeatures
    monkeypatch.setattr(hip, "_lib", None)
    monkeypatch.setattr(hip, "_LIB_PATH", "/nonexistent/libkeystone_hip.so")
    monkeypatch.delenv("KEYSTONE_ALLOW_EAGER", raising=False)
    op = CosineRandomFeatures(8, 16, gamma=1.0)
    X = torch.randn(4, 8, device="cuda", dtype=torch.float32)
    import pytest as _pytest
    with _pytest.raises(RuntimeError, match="refusing"):
        op.apply_batch(X)
    monkeypatch.setenv("KEYSTONE_ALLOW_EAGER", "1")
    out = op.apply_batch(X)          # authorized eager fallback
    assert out.shape == (4, 16)
