"""Vocabulary selection and sparse vectorization.

ref: nodes.misc.CommonSparseFeatures (top-K features by
document frequency -> sparse vectorizer index map), SparseFeatureVectorizer,
nodes.nlp.WordFrequencyEncoder (word -> frequency-rank index).

The fitted vectorizer emits torch sparse CSR shards — the bridge from
CPU-side text preprocessing to the GPU solvers (SparseLBFGS etc.).
"""

from __future__ import annotations

from collections import Counter
from typing import Dict, List, Sequence

import torch

from ...parallel.dataset import Dataset, ItemDataset, TensorDataset
from ...workflow.pipeline import Estimator, Transformer


def _merge_counters(local: Counter, runtime) -> Counter:
    if not runtime.is_distributed:
        return local
    import torch.distributed as dist
    gathered = [None] * runtime.world_size
    dist.all_gather_object(gathered, local)
    total = Counter()
    for c in gathered:
        total.update(c)
    return total


class SparseFeatureVectorizer(Transformer):
    """{term: weight} -> sparse row under a fixed term->index map."""

    def __init__(self, index: Dict, num_features: int):
        self.index = index
        self.num_features = num_features

    def apply(self, tf: Dict) -> torch.Tensor:
        idx, val = [], []
        for t, v in tf.items():
            j = self.index.get(t)
            if j is not None:
                idx.append(j)
                val.append(v)
        out = torch.zeros(self.num_features)
        if idx:
            out[torch.tensor(idx)] = torch.tensor(val, dtype=torch.float32)
        return out

    def apply_dataset(self, ds: Dataset) -> TensorDataset:
        rows, cols, vals = [], [], []
        n = 0
        for tf in ds.items():
            for t, v in tf.items():
                j = self.index.get(t)
                if j is not None:
                    rows.append(n)
                    cols.append(j)
                    vals.append(float(v))
            n += 1
        i = torch.tensor([rows, cols], dtype=torch.int64)
        v = torch.tensor(vals, dtype=torch.float32)
        dense_shape = (n, self.num_features)
        csr = torch.sparse_coo_tensor(i, v, dense_shape).coalesce().to_sparse_csr()
        return TensorDataset(csr.to(ds.runtime.device), ds.runtime)

    def get_config(self):
        # keys may be NGram tuples (n-gram terms) or plain strings; encode
        # tuples as lists so the load path rebuilds equal-hashing keys
        terms = [list(t) if isinstance(t, tuple) else t for t in self.index]
        return {"num_features": self.num_features, "terms": terms,
                "indices": list(self.index.values())}

    @classmethod
    def from_saved(cls, config, state):
        from .ngrams import NGram
        index = {}
        for t, i in zip(config["terms"], config["indices"]):
            key = NGram(t) if isinstance(t, list) else t
            index[key] = i
        return cls(index, config["num_features"])


class CommonSparseFeatures(Estimator):
    """Keep the ``num_features`` terms with highest document frequency
    (ref: CommonSparseFeatures(numFeatures)).  Deterministic tie-break by
    term string so all ranks build the identical map."""

    def __init__(self, num_features: int):
        self.num_features = num_features

    def _fit(self, data: ItemDataset) -> SparseFeatureVectorizer:
        df = Counter()
        for tf in data.items():
            df.update(tf.keys())
        df = _merge_counters(df, data.runtime)
        top = sorted(df.items(), key=lambda kv: (-kv[1], str(kv[0])))
        top = top[: self.num_features]
        index = {t: i for i, (t, _c) in enumerate(top)}
        return SparseFeatureVectorizer(index, self.num_features)


class WordFrequencyEncoder(Estimator):
    """word -> frequency rank (0 = most frequent); unseen -> -1
    (ref: WordFrequencyEncoder)."""

    def _fit(self, data: ItemDataset) -> "WordFrequencyEncoderModel":
        counts = Counter()
        for tokens in data.items():
            counts.update(tokens)
        counts = _merge_counters(counts, data.runtime)
        ranked = sorted(counts.items(), key=lambda kv: (-kv[1], kv[0]))
        return WordFrequencyEncoderModel({w: i for i, (w, _c) in enumerate(ranked)})


class WordFrequencyEncoderModel(Transformer):
    def __init__(self, ranks: Dict[str, int]):
        self.ranks = ranks

    def apply(self, tokens: Sequence[str]) -> List[int]:
        return [self.ranks.get(t, -1) for t in tokens]

    def get_config(self):
        items = sorted(self.ranks.items(), key=lambda kv: kv[1])
        return {"words": [w for w, _ in items]}

    @classmethod
    def from_saved(cls, config, state):
        return cls({w: i for i, w in enumerate(config["words"])})
