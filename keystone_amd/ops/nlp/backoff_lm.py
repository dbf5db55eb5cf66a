"""Stupid Backoff n-gram language model (Brants et al. 2007).

ref: nodes.nlp.StupidBackoffEstimator + the bit-packed
NGramIndexer / NaiveBitPackIndexer.  Scores
``S(w | context) = count(context + w) / count(context)`` backing off with
factor alpha=0.4 per dropped context word; unigram base
``count(w) / N``.  Counts are bit-packed int64 keys (21 bits per word id,
up to 3-gram) in plain dicts; corpus counts are merged across ranks.
"""

from __future__ import annotations

import math
from collections import Counter
from typing import Dict, List, Sequence

from ...parallel.dataset import ItemDataset
from ...workflow.pipeline import Estimator, Transformer
from .vocab import _merge_counters

PACK_BITS = 21
PACK_MASK = (1 << PACK_BITS) - 1


class NGramIndexer:
    """Bit-pack up to 3 word ids into one int64 key (ref: NaiveBitPackIndexer)."""

    max_order = 3

    @staticmethod
    def pack(ids: Sequence[int]) -> int:
        assert 1 <= len(ids) <= NGramIndexer.max_order
        key = len(ids)
        for w in ids:
            assert 0 <= w <= PACK_MASK, "word id exceeds 21 bits"
            key = (key << PACK_BITS) | w
        return key

    @staticmethod
    def unpack(key: int) -> List[int]:
        ids = []
        tmp = key
        while tmp > (1 << PACK_BITS):
            ids.append(tmp & PACK_MASK)
            tmp >>= PACK_BITS
        return list(reversed(ids))


class StupidBackoffModel(Transformer):
    def __init__(self, counts: Dict[int, int], vocab: Dict[str, int],
                 total_tokens: int, alpha: float = 0.4, max_order: int = 3):
        self.counts = counts
        self.vocab = vocab
        self.total_tokens = total_tokens
        self.alpha = alpha
        self.max_order = max_order

    def _score_ids(self, context: List[int], w: int) -> float:
        """S(w | context) with backoff; returns a probability-like score."""
        for drop in range(len(context) + 1):
            ctx = context[drop:]
            if ctx:
                num = self.counts.get(NGramIndexer.pack(ctx + [w]), 0)
                den = self.counts.get(NGramIndexer.pack(ctx), 0)
                if num > 0 and den > 0:
                    return (self.alpha ** drop) * num / den
            else:
                num = self.counts.get(NGramIndexer.pack([w]), 0)
                return (self.alpha ** drop) * max(num, 0.4) / self.total_tokens
        return 1e-10

    def apply(self, tokens: Sequence[str]) -> float:
        """Mean log-score of the token sequence under the LM."""
        ids = [self.vocab.get(t, -1) for t in tokens]
        lp = 0.0
        for i, w in enumerate(ids):
            if w < 0:
                lp += math.log(1e-10)
                continue
            ctx = [c for c in ids[max(0, i - self.max_order + 1): i] if c >= 0]
            lp += math.log(max(self._score_ids(ctx, w), 1e-300))
        return lp / max(len(ids), 1)

    def get_config(self):
        items = sorted(self.vocab.items(), key=lambda kv: kv[1])
        return {"words": [w for w, _ in items],
                "keys": list(self.counts.keys()),
                "vals": list(self.counts.values()),
                "total_tokens": self.total_tokens,
                "alpha": self.alpha, "max_order": self.max_order}

    @classmethod
    def from_saved(cls, config, state):
        return cls(dict(zip(config["keys"], config["vals"])),
                   {w: i for i, w in enumerate(config["words"])},
                   config["total_tokens"], config["alpha"],
                   config["max_order"])


class StupidBackoffEstimator(Estimator):
    def __init__(self, max_order: int = 3, alpha: float = 0.4):
        assert max_order <= NGramIndexer.max_order
        self.max_order = max_order
        self.alpha = alpha

    def _fit(self, data: ItemDataset) -> StupidBackoffModel:
        # build the vocab (frequency-ranked, deterministic)
        wc = Counter()
        for tokens in data.items():
            wc.update(tokens)
        wc = _merge_counters(wc, data.runtime)
        ranked = sorted(wc.items(), key=lambda kv: (-kv[1], kv[0]))
        vocab = {w: i for i, (w, _c) in enumerate(ranked)}
        total = sum(wc.values())

        counts: Counter = Counter()
        for tokens in data.items():
            ids = [vocab[t] for t in tokens]
            for n in range(1, self.max_order + 1):
                for i in range(len(ids) - n + 1):
                    counts[NGramIndexer.pack(ids[i:i + n])] += 1
        counts = _merge_counters(counts, data.runtime)
        return StupidBackoffModel(dict(counts), vocab, total, self.alpha,
                                  self.max_order)
