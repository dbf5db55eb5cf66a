"""Text pipelines: 20 Newsgroups and Amazon Reviews.

ref: pipelines/text/NewsgroupsPipeline and
pipelines/text/AmazonReviewsPipeline — driver configs 1 and 4.

  Newsgroups: Trim -> LowerCase -> Tokenizer -> NGrams(1..3) -> TF ->
              CommonSparseFeatures(100k) -> NaiveBayes -> MaxClassifier
  Amazon:     same featurization (1..2-grams, sqrt TF) ->
              LogisticRegression (binary)
"""

from __future__ import annotations

from dataclasses import dataclass

from ..ops.learning import LogisticRegressionEstimator, NaiveBayesEstimator
from ..ops.nlp import (CommonSparseFeatures, LowerCase, NGramsFeaturizer,
                       TermFrequency, Tokenizer, Trim)
from ..ops.util import MaxClassifier
from ..workflow.pipeline import Pipeline


@dataclass
class NewsgroupsConfig:
    num_classes: int = 20
    ngrams: int = 3
    common_features: int = 100000


def newsgroups_featurizer(conf: NewsgroupsConfig):
    return (Trim().and_then(LowerCase()).and_then(Tokenizer())
            .and_then(NGramsFeaturizer(1, conf.ngrams))
            .and_then(TermFrequency("identity")))


def build_newsgroups(train_docs, train_labels,
                     conf: NewsgroupsConfig = NewsgroupsConfig()) -> Pipeline:
    feats = newsgroups_featurizer(conf)
    return (feats
            .and_then(CommonSparseFeatures(conf.common_features), train_docs)
            .and_then(NaiveBayesEstimator(conf.num_classes), train_docs,
                      train_labels)
            .and_then(MaxClassifier()))


@dataclass
class AmazonConfig:
    ngrams: int = 2
    common_features: int = 100000
    num_iters: int = 20
    reg: float = 1e-5


def build_amazon(train_docs, train_labels,
                 conf: AmazonConfig = AmazonConfig()) -> Pipeline:
    feats = (Trim().and_then(LowerCase()).and_then(Tokenizer())
             .and_then(NGramsFeaturizer(1, conf.ngrams))
             .and_then(TermFrequency("sqrt")))
    return (feats
            .and_then(CommonSparseFeatures(conf.common_features), train_docs)
            .and_then(LogisticRegressionEstimator(
                num_classes=2, num_iters=conf.num_iters, reg_param=conf.reg),
                train_docs, train_labels))
