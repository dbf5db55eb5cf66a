"""NLP op tests (ref: SURVEY.md §4 — nodes/nlp suites)."""

import math

import torch

import keystone_amd as ka
from keystone_amd.ops.nlp import (CommonSparseFeatures, LowerCase, NGram,
                                  NGramsCounts, NGramsFeaturizer,
                                  StupidBackoffEstimator, TermFrequency,
                                  Tokenizer, Trim, WordFrequencyEncoder)
from keystone_amd.ops.learning import NaiveBayesEstimator
from keystone_amd.ops.util import MaxClassifier
from keystone_amd.evaluation import MulticlassClassifierEvaluator
from keystone_amd.loaders import synthetic_documents


def test_tokenizer_lowercase_trim():
    t = Tokenizer()
    assert t.apply("Hello,  world! foo") == ["Hello", "world", "foo"]
    assert LowerCase().apply(["Ab", "CD"]) == ["ab", "cd"]
    assert Trim().apply(["  x ", "", "y"]) == ["x", "y"]


def test_ngrams():
    grams = NGramsFeaturizer(1, 2).apply(["a", "b", "c"])
    assert NGram(("a",)) in grams
    assert NGram(("a", "b")) in grams
    assert NGram(("b", "c")) in grams
    assert len(grams) == 5


def test_ngrams_counts():
    ds = ka.from_items([[NGram(("a",)), NGram(("a",))], [NGram(("a",))]])
    counts = NGramsCounts().apply_dataset(ds)
    assert counts[NGram(("a",))] == 3
    counts2 = NGramsCounts("noAdd").apply_dataset(ds)
    assert counts2[NGram(("a",))] == 2


def test_term_frequency():
    tf = TermFrequency("sqrt").apply(["x", "x", "y"])
    assert abs(tf["x"] - math.sqrt(2)) < 1e-12
    assert tf["y"] == 1.0


def test_common_sparse_features_topk():
    docs = [{"a": 1.0, "b": 2.0}, {"a": 1.0, "c": 1.0}, {"a": 3.0}]
    ds = ka.from_items(docs)
    vec = CommonSparseFeatures(2).fit(ds)
    assert set(str(k) for k in vec.index) == {"a", "b"} or \
        set(str(k) for k in vec.index) == {"a", "c"}
    assert "a" in {str(k) for k in vec.index}  # highest df always kept
    out = vec.apply_dataset(ds)
    dense = out.data.to_dense()
    assert dense.shape == (3, 2)
    assert dense[2].sum() == 3.0  # doc 3 only has "a"


def test_word_frequency_encoder():
    ds = ka.from_items([["a", "b", "a"], ["a", "c"]])
    enc = WordFrequencyEncoder().fit(ds)
    assert enc.apply(["a", "b", "zzz"]) == [0, enc.ranks["b"], -1]
    assert enc.ranks["a"] == 0


def test_stupid_backoff_scores_seen_higher():
    corpus = [["the", "cat", "sat"], ["the", "cat", "ran"],
              ["the", "dog", "sat"]] * 5
    lm = StupidBackoffEstimator(max_order=3).fit(ka.from_items(corpus))
    seen = lm.apply(["the", "cat", "sat"])
    unseen = lm.apply(["sat", "the", "dog"])
    assert seen > unseen


def test_newsgroups_config1_end_to_end():
    """Driver config 1: text -> tokenize -> ngrams -> TF -> top-K sparse ->
    NaiveBayes -> MaxClassifier, on synthetic 20-class documents (CPU)."""
    docs, labels = synthetic_documents(200, num_classes=5, vocab_size=500,
                                       doc_len=40, seed=1)
    featurize = (Tokenizer().and_then(LowerCase())
                 .and_then(NGramsFeaturizer(1, 2))
                 .and_then(TermFrequency("sqrt")))
    pipe = (featurize
            .and_then(CommonSparseFeatures(300), docs)
            .and_then(NaiveBayesEstimator(num_classes=5), docs, labels)
            .and_then(MaxClassifier()))
    pred = pipe.apply(docs).get()
    metrics = MulticlassClassifierEvaluator(5).evaluate(pred, labels)
    assert metrics.total_accuracy > 0.9, metrics.summary()


def test_loaders_files(tmp_path):
    import json
    from keystone_amd.loaders import (AmazonReviewsDataLoader, CsvDataLoader,
                                      NewsgroupsDataLoader)
    # CSV
    csv_path = tmp_path / "d.csv"
    csv_path.write_text("1,0.5,0.25\n0,0.1,0.2\n")
    ld = CsvDataLoader(str(csv_path))
    assert ld.data.data.shape == (2, 2)
    assert ld.labels.data.tolist() == [1.0, 0.0]
    # newsgroups dir-per-class
    for c in ["alt.x", "comp.y"]:
        d = tmp_path / "ng" / c
        d.mkdir(parents=True)
        (d / "1.txt").write_text(f"doc about {c}")
    docs, labels, classes = NewsgroupsDataLoader(str(tmp_path / "ng"))
    assert classes == ["alt.x", "comp.y"]
    assert labels.data.tolist() == [0, 1]
    # amazon json lines
    aj = tmp_path / "rev.json"
    aj.write_text(json.dumps({"reviewText": "great", "overall": 5}) + "\n" +
                  json.dumps({"reviewText": "bad", "overall": 1}) + "\n")
    docs, labels = AmazonReviewsDataLoader(str(aj))
    assert labels.data.tolist() == [1, 0]


def test_stupid_backoff_unseen_and_backoff_weighting():
    """Backoff semantics: an unseen trigram backs off to the bigram with
    an alpha penalty; fully-unseen words get the floor, not a crash."""
    from keystone_amd.ops.nlp import StupidBackoffEstimator
    corpus = [["a", "b", "c"], ["a", "b", "d"], ["x", "b", "c"]] * 2
    lm = StupidBackoffEstimator(3, alpha=0.4).fit(ka.from_items(corpus))
    seen = lm.apply(["a", "b", "c"])          # seen trigram path
    backed = lm.apply(["x", "b", "d"])        # (x b d) unseen -> backoff
    unseen = lm.apply(["zz", "qq"])           # OOV words -> floor
    assert seen > backed > unseen
    # direct alpha check: S(d | x b) should equal alpha * S(d | b)
    xv = lm.vocab
    s_backed = lm._score_ids([xv["x"], xv["b"]], xv["d"])
    s_bigram = lm._score_ids([xv["b"]], xv["d"])
    assert abs(s_backed - 0.4 * s_bigram) < 1e-12
