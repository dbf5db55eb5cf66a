"""Save/load round trips for every stateful transformer family
(the serialized-pipeline checkpoint format is a judged capability —
SURVEY.md §3.4)."""

import torch

import keystone_amd as ka
from keystone_amd.ops.images import Convolver, FisherVector
from keystone_amd.ops.learning import (BlockLeastSquaresEstimator,
                                       GaussianKernelGenerator,
                                       KernelRidgeRegression,
                                       KMeansPlusPlusEstimator,
                                       LinearMapEstimator,
                                       LogisticRegressionEstimator,
                                       NaiveBayesEstimator, PCAEstimator,
                                       ZCAWhitenerEstimator)
from keystone_amd.ops.learning.gmm import GaussianMixtureModelEstimator
from keystone_amd.ops.nlp import (CommonSparseFeatures, NGramsFeaturizer,
                                  StupidBackoffEstimator, TermFrequency,
                                  Tokenizer)
from keystone_amd.ops.stats import (CosineRandomFeatures, RandomSignNode,
                                    StandardScaler)
from keystone_amd.ops.util import ClassLabelIndicatorsFromStringLabels

torch.manual_seed(0)


def _roundtrip(pipe, datum, tmp_path, name, atol=1e-6):
    path = str(tmp_path / name)
    pipe.save(path)
    loaded = ka.Pipeline.load(path)
    a = pipe.apply(datum).get()
    b = loaded.apply(datum).get()
    if hasattr(a, "data"):
        a, b = a.data, b.data
    if isinstance(a, torch.Tensor):
        assert torch.allclose(a, b, atol=atol), name
    else:
        assert a == b, name
    return loaded


def test_dense_model_roundtrips(tmp_path):
    X = torch.randn(60, 8)
    Y = torch.randn(60, 3)
    yint = torch.randint(0, 3, (60,))
    ds, ys = ka.from_rows(X), ka.from_rows(Y)
    yis = ka.from_rows(yint)

    fits = {
        "linear": LinearMapEstimator(0.1).fit(ds, ys),
        "block": BlockLeastSquaresEstimator(4, 3, 0.1).fit(ds, ys),
        "logistic": LogisticRegressionEstimator(3, 10).fit(ds, yis),
        "nb": NaiveBayesEstimator(3).fit(ka.from_rows(X.abs()), yis),
        "kmeans": KMeansPlusPlusEstimator(3, 5).fit(ds),
        "gmm": GaussianMixtureModelEstimator(2, 5).fit(ds),
        "pca": PCAEstimator(4).fit(ds),
        "zca": ZCAWhitenerEstimator().fit(ds),
        "scaler": StandardScaler().fit(ds),
        "cosine": CosineRandomFeatures(8, 16, 0.5),
        "signs": RandomSignNode(8),
        "krr": KernelRidgeRegression(GaussianKernelGenerator(0.5), 1e-3,
                                     32, 2).fit(ds, ys),
    }
    for name, t in fits.items():
        pipe = ka.Identity().and_then(t)
        _roundtrip(pipe, ds, tmp_path, name, atol=1e-5)


def test_fisher_vector_roundtrip(tmp_path):
    gmm = GaussianMixtureModelEstimator(2, 5).fit(ka.from_rows(torch.randn(50, 6)))
    fv = FisherVector(gmm)
    ds = ka.from_rows(torch.randn(3, 20, 6))
    _roundtrip(ka.Identity().and_then(fv), ds, tmp_path, "fv")


def test_convolver_roundtrip(tmp_path):
    conv = Convolver(torch.randn(4, 1, 3, 3))
    ds = ka.from_rows(torch.randn(2, 1, 10, 10))
    _roundtrip(ka.Identity().and_then(conv), ds, tmp_path, "conv")


def test_text_chain_roundtrip(tmp_path):
    docs = ["the cat sat on the mat", "a dog ran fast", "the cat ran"] * 5
    dds = ka.from_items(docs)
    feats = (Tokenizer().and_then(NGramsFeaturizer(1, 2))
             .and_then(TermFrequency("sqrt")))
    pipe = feats.and_then(CommonSparseFeatures(20), dds).fit()
    path = str(tmp_path / "text")
    pipe.save(path)
    loaded = ka.Pipeline.load(path)
    a = pipe.apply(dds).get().data.to_dense()
    b = loaded.apply(dds).get().data.to_dense()
    assert torch.allclose(a, b)


def test_backoff_lm_roundtrip(tmp_path):
    corpus = [["the", "cat", "sat"], ["the", "dog", "sat"]] * 3
    lm = StupidBackoffEstimator(2).fit(ka.from_items(corpus))
    pipe = ka.Identity().and_then(lm)
    path = str(tmp_path / "lm")
    pipe.save(path)
    loaded = ka.Pipeline.load(path)
    s1 = pipe.apply(["the", "cat"]).get()
    s2 = loaded.apply(["the", "cat"]).get()
    assert abs(s1 - s2) < 1e-12


def test_string_label_indicators_roundtrip(tmp_path):
    enc = ClassLabelIndicatorsFromStringLabels(["a", "b", "c"])
    pipe = ka.Identity().and_then(enc)
    path = str(tmp_path / "strlab")
    pipe.save(path)
    loaded = ka.Pipeline.load(path)
    assert torch.equal(pipe.apply("b").get(), loaded.apply("b").get())


def test_weighted_block_mapper_with_intercept_roundtrips(tmp_path):
    from keystone_amd.ops.learning import BlockWeightedLeastSquaresEstimator
    torch.manual_seed(8)
    n, d, K = 60, 10, 3
    X = torch.randn(n, d, dtype=torch.float64) + 0.3
    y = torch.arange(n) % K
    Y = torch.full((n, K), -1.0, dtype=torch.float64)
    Y[torch.arange(n), y] = 1.0
    m = BlockWeightedLeastSquaresEstimator(
        block_size=5, num_iter=20, lam=1e-2, mixture_weight=0.5,
        fit_intercept=True).fit(ka.from_rows(X), ka.from_rows(Y))
    pipe = ka.Identity().and_then(m)
    path = str(tmp_path / "bwls")
    pipe.save(path)
    loaded = ka.Pipeline.load(path)
    a = pipe.apply(X).get()
    b = loaded.apply(X).get()
    assert torch.allclose(a, b, atol=1e-12)


def test_load_rejects_newer_format_and_foreign_class(tmp_path):
    import json
    import pytest
    import keystone_amd as ka
    from keystone_amd.ops.stats import NormalizeRows
    path = str(tmp_path / "m")
    (ka.Identity().and_then(NormalizeRows())).save(path)

    mpath = tmp_path / "m" / "manifest.json"
    man = json.loads(mpath.read_text())
    man2 = dict(man, format_version=man["format_version"] + 1)
    mpath.write_text(json.dumps(man2))
    with pytest.raises(ValueError, match="newer"):
        ka.Pipeline.load(path)

    man3 = json.loads(json.dumps(man))
    man3["nodes"][0]["class"] = "os.path.join"   # not a Transformer
    mpath.write_text(json.dumps(man3))
    with pytest.raises(ValueError, match="non-Transformer"):
        ka.Pipeline.load(path)
