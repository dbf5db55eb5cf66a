"""End-to-end pipeline tests for the five driver configs (small CPU sizes).
ref: SURVEY.md §4 — the driver configs are the integration benchmarks."""

import os
import pytest
import torch

import keystone_amd as ka
from keystone_amd.evaluation import (BinaryClassifierEvaluator,
                                     MeanAveragePrecisionEvaluator,
                                     MulticlassClassifierEvaluator)
from keystone_amd.loaders import (synthetic_dense, synthetic_documents,
                                  synthetic_images)
from keystone_amd.models import (AmazonConfig, CifarConfig, ImageNetFVConfig,
                                 MnistConfig, TimitConfig, VOCConfig,
                                 build_amazon, build_mnist_randomfft,
                                 build_newsgroups, build_timit,
                                 fit_cifar_patches, fit_imagenet_fv, fit_voc,
                                 NewsgroupsConfig)

torch.manual_seed(0)


def test_config1_newsgroups():
    docs, labels = synthetic_documents(150, num_classes=4, vocab_size=400,
                                       doc_len=30, seed=2)
    pipe = build_newsgroups(docs, labels, NewsgroupsConfig(
        num_classes=4, ngrams=2, common_features=400))
    pred = pipe.apply(docs).get()
    m = MulticlassClassifierEvaluator(4).evaluate(pred, labels)
    assert m.total_accuracy > 0.9


def test_config2_mnist_randomfft():
    train, labels = synthetic_dense(300, 64, 10, seed=3, noise=0.8)
    pipe = build_mnist_randomfft(train, labels, MnistConfig(
        input_dim=64, num_ffts=2, num_classes=10, block_size=32, num_iter=3))
    pred = pipe.apply(train).get()
    m = MulticlassClassifierEvaluator(10).evaluate(pred, labels)
    assert m.total_accuracy > 0.9


def test_config3_timit():
    train, labels = synthetic_dense(400, 80, 12, seed=4, noise=1.0)
    pipe = build_timit(train, labels, TimitConfig(
        input_dim=80, num_features=512, gamma=0.02, num_classes=12,
        block_size=128, num_iter=3, lam=1e-4))
    pred = pipe.apply(train).get()
    m = MulticlassClassifierEvaluator(12).evaluate(pred, labels)
    assert m.total_accuracy > 0.85


def test_config4_amazon():
    docs, labels = synthetic_documents(200, num_classes=2, vocab_size=300,
                                       doc_len=40, seed=5)
    pipe = build_amazon(docs, labels, AmazonConfig(
        ngrams=2, common_features=300, num_iters=15))
    pred = pipe.apply(docs).get()
    m = BinaryClassifierEvaluator().evaluate(pred, labels)
    assert m.accuracy > 0.9


def test_config5_imagenet_fv_small():
    imgs, labels = synthetic_images(12, h=48, w=48, num_classes=4, seed=6)
    conf = ImageNetFVConfig(num_classes=4, sift_step=6, sift_bin=4,
                            sift_scales=1, lcs_step=6, lcs_subpatch=4,
                            pca_dims=16, gmm_k=4, n_sample=2000,
                            block_size=128, solver_epochs=2, lam=1e-4,
                            top_k=1)
    pipe, info = fit_imagenet_fv(imgs, labels, conf)
    assert info["feature_dim"] == 2 * 4 * 16 * 2  # 2KD per branch, 2 branches
    pred = pipe.apply(imgs).get().data.squeeze(1)
    # tiny synthetic set: the solver should at least fit its own train set
    acc = (pred == labels.data).float().mean().item()
    assert acc > 0.5


def test_cifar_patches_pipeline():
    imgs = ka.from_rows(torch.rand(40, 3, 16, 16))
    labels = ka.from_rows(torch.randint(0, 3, (40,)))
    pipe = fit_cifar_patches(imgs, labels, CifarConfig(
        num_filters=16, patch_size=4, patch_sample=500, pool_size=6,
        pool_stride=6, lam=1.0, num_classes=3))
    pred = pipe.apply(imgs).get()
    m = MulticlassClassifierEvaluator(3).evaluate(pred, labels)
    assert m.total_accuracy > 0.5  # memorize small train set


def test_voc_map_pipeline():
    imgs, _ = synthetic_images(10, h=40, w=40, num_classes=3, seed=7)
    Y = (torch.rand(imgs.data.shape[0], 3) > 0.5).float()
    Y[:, 0] = 1.0  # ensure a positive class
    labels = ka.from_rows(Y)
    pipe = fit_voc(imgs, labels, VOCConfig(
        num_classes=3, sift_step=6, sift_bin=4, sift_scales=1, pca_dims=8,
        gmm_k=2, n_sample=1000, block_size=16, solver_epochs=2))
    scores = pipe.apply(imgs).get()
    res = MeanAveragePrecisionEvaluator(3).evaluate(scores, labels)
    assert 0.0 <= res["mAP"] <= 1.0
    assert res["mAP"] > 0.5  # train-set ranking should be decent


def test_config2_mnist_random_cosine():
    from keystone_amd.models import build_mnist_random_cosine
    train, labels = synthetic_dense(300, 64, 10, seed=8, noise=0.8)
    pipe = build_mnist_random_cosine(train, labels, MnistConfig(
        input_dim=64, num_classes=10, block_size=128, num_iter=3, lam=1e-4),
        num_features=512, gamma=0.02)
    pred = pipe.apply(train).get()
    m = MulticlassClassifierEvaluator(10).evaluate(pred, labels)
    assert m.total_accuracy > 0.9


def test_config5_imagenet_fv_color():
    imgs, labels = synthetic_images(10, h=48, w=48, num_classes=4, seed=9,
                                    channels=3)
    assert imgs.data.dim() == 4
    conf = ImageNetFVConfig(num_classes=4, sift_step=6, sift_bin=4,
                            sift_scales=1, lcs_step=6, lcs_subpatch=4,
                            pca_dims=16, gmm_k=4, n_sample=2000,
                            block_size=128, solver_epochs=2, lam=1e-4,
                            top_k=1)
    pipe, info = fit_imagenet_fv(imgs, labels, conf)
    # LCS branch sees color: 96-dim raw descriptors -> PCA 16
    pred = pipe.apply(imgs).get().data.squeeze(1)
    acc = (pred == labels.data).float().mean().item()
    assert acc > 0.5


def test_imagenet_fv_generalizes_on_structured_classes():
    """Statistical validation of the full featurization+solver stack:
    class-structured synthetic images, disjoint train/test, accuracy must
    beat chance by a wide margin (ref test strategy: accuracy-parity
    integration checks)."""
    K = 4
    train, ytr = synthetic_images(96, h=64, w=64, num_classes=K, seed=31,
                                  class_structure=True)
    test, yte = synthetic_images(48, h=64, w=64, num_classes=K, seed=77,
                                 class_structure=True)
    conf = ImageNetFVConfig(num_classes=K, sift_step=4, sift_bin=4,
                            sift_scales=1, lcs_step=4, lcs_subpatch=4,
                            pca_dims=16, gmm_k=8, n_sample=8000,
                            block_size=128, solver_epochs=3, lam=0.1,
                            top_k=1, gmm_iters=10)
    pipe, _info = fit_imagenet_fv(train, ytr, conf)
    pred = pipe.apply(test).get().data.squeeze(1)
    acc = (pred == yte.data).float().mean().item()
    # raw-pixel nearest-centroid scores ~0.08 on this task (position-
    # variant); >= 0.45 at chance 0.25 over 48 samples is p << 1e-3 —
    # the FV stack genuinely generalizes (measured ~0.58)
    assert acc > 0.45, f"test accuracy {acc} vs chance {1/K}"


def test_cli_cifar_and_voc(tmp_path):
    """The two remaining reference example apps run end-to-end from the
    CLI (ref L6: RandomPatchCifar, VOCSIFTFisher)."""
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "bin/run_pipeline.py", "cifar", "--n", "300",
         "--num-filters", "32"], cwd=root, capture_output=True, text=True,
        timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "train accuracy" in out.stdout
    out = subprocess.run(
        [sys.executable, "bin/run_pipeline.py", "voc", "--images", "16",
         "--image-size", "64"], cwd=root, capture_output=True, text=True,
        timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "train mAP" in out.stdout


def test_mean_ap_hand_computed():
    """AP against a worked example, plain and 11-point interpolated."""
    # one class, ranked relevance pattern [1, 0, 1, 0]:
    # plain AP = (1/1 + 2/3) / 2 = 5/6
    scores = torch.tensor([[0.9], [0.8], [0.7], [0.1]])
    acts = torch.tensor([[1.0], [0.0], [1.0], [0.0]])
    ev = MeanAveragePrecisionEvaluator(1)
    res = ev.evaluate(ka.from_rows(scores), ka.from_rows(acts))
    assert abs(res["mAP"] - 5.0 / 6.0) < 1e-9
    # 11-pt: recall thresholds 0..1; max precision at recall>=t is 1.0
    # for t <= 0.5 (6 thresholds) and 2/3 for t > 0.5 (5 thresholds)
    ev11 = MeanAveragePrecisionEvaluator(1, interpolated_11pt=True)
    res11 = ev11.evaluate(ka.from_rows(scores), ka.from_rows(acts))
    expect = (6 * 1.0 + 5 * (2.0 / 3.0)) / 11
    assert abs(res11["mAP"] - expect) < 1e-9


def test_classifier_metrics_hand_oracle():
    """Confusion / per-class / macro-F1 / binary PRF against hand-counted
    values on a tiny fixed prediction set."""
    import keystone_amd as ka
    from keystone_amd.evaluation import (BinaryClassifierEvaluator,
                                         MulticlassClassifierEvaluator)
    act = torch.tensor([0, 0, 0, 1, 1, 2])
    pred = torch.tensor([0, 1, 0, 1, 2, 2])
    m = MulticlassClassifierEvaluator(3).evaluate(
        ka.from_rows(pred), ka.from_rows(act))
    # confusion rows=actual: [[2,1,0],[0,1,1],[0,0,1]]
    assert m.confusion.tolist() == [[2, 1, 0], [0, 1, 1], [0, 0, 1]]
    assert abs(m.total_accuracy - 4 / 6) < 1e-12
    assert torch.allclose(m.per_class_accuracy,
                          torch.tensor([2 / 3, 1 / 2, 1.0],
                                       dtype=torch.float64))
    # per-class precision: [1, 1/2, 1/2]; recall: [2/3, 1/2, 1]
    f1 = [2 * 1 * (2 / 3) / (1 + 2 / 3), 2 * 0.5 * 0.5 / 1.0,
          2 * 0.5 * 1 / 1.5]
    assert abs(m.macro_f1 - sum(f1) / 3) < 1e-12

    b_act = torch.tensor([1, 1, 1, 0, 0, 0])
    b_pred = torch.tensor([1, 0, 1, 1, 0, 0])
    bm = BinaryClassifierEvaluator().evaluate(
        ka.from_rows(b_pred), ka.from_rows(b_act))
    assert (bm.tp, bm.fp, bm.tn, bm.fn) == (2, 1, 2, 1)
    assert abs(bm.precision - 2 / 3) < 1e-12
    assert abs(bm.recall - 2 / 3) < 1e-12
    assert abs(bm.f1 - 2 / 3) < 1e-12


def test_imagenet_fv_solver_checkpoint_resume(tmp_path):
    """Flagship-API failure recovery: interrupt after 2 solver epochs,
    resume to 4 via solver_checkpoint_dir — identical model to the
    uninterrupted 4-epoch fit (featurization is deterministic per seed)."""
    import torch
    from keystone_amd.loaders.synthetic import synthetic_images
    from keystone_amd.models.imagenet_fv import (ImageNetFVConfig,
                                                 fit_imagenet_fv)
    imgs, labels = synthetic_images(16, h=48, w=48, num_classes=3, seed=3)
    kw = dict(num_classes=3, sift_step=6, sift_bin=4, sift_scales=1,
              lcs_step=6, lcs_subpatch=4, pca_dims=8, gmm_k=4,
              n_sample=2000, block_size=16, lam=0.1, top_k=1, gmm_iters=3)
    _, full = fit_imagenet_fv(imgs, labels,
                              ImageNetFVConfig(solver_epochs=4, **kw))
    ck = str(tmp_path / "solver_ck")
    fit_imagenet_fv(imgs, labels, ImageNetFVConfig(
        solver_epochs=2, solver_checkpoint_dir=ck, **kw))
    _, resumed = fit_imagenet_fv(imgs, labels, ImageNetFVConfig(
        solver_epochs=4, solver_checkpoint_dir=ck, **kw))
    W_full = torch.cat(list(full["model"].blocks), dim=0)
    W_res = torch.cat(list(resumed["model"].blocks), dim=0)
    assert torch.allclose(W_full, W_res, atol=1e-5), \
        (W_full - W_res).abs().max()
