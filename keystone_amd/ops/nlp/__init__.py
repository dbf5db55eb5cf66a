from .tokenizer import LowerCase, Tokenizer, Trim
from .ngrams import NGram, NGramsFeaturizer, NGramsCounts
from .termfreq import TermFrequency
from .vocab import (CommonSparseFeatures, SparseFeatureVectorizer,
                    WordFrequencyEncoder)
from .backoff_lm import NGramIndexer, StupidBackoffEstimator, StupidBackoffModel

__all__ = [
    "Tokenizer", "LowerCase", "Trim", "NGram", "NGramsFeaturizer",
    "NGramsCounts", "TermFrequency", "CommonSparseFeatures",
    "SparseFeatureVectorizer", "WordFrequencyEncoder",
    "StupidBackoffEstimator", "StupidBackoffModel", "NGramIndexer",
]
