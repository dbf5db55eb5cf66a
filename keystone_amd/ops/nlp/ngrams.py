"""N-gram featurization and corpus counting.

ref: nodes.nlp.NGramsFeaturizer (token seq -> all n-grams
for n in [lo, hi]), NGram, NGramsCounts (corpus-level counting via Spark
``reduceByKey``).  The distributed count merges local dicts with an
all-gather-object exchange (host-side; n-gram keys are strings).
"""

from __future__ import annotations

from collections import Counter
from typing import List, Sequence

from ...parallel.dataset import Dataset
from ...workflow.pipeline import Transformer


class NGram(tuple):
    """An n-gram: a tuple of tokens (ref: NGram value class)."""
    __slots__ = ()

    def __str__(self):
        return " ".join(self)


class NGramsFeaturizer(Transformer):
    def __init__(self, lo: int = 1, hi: int = 2):
        self.lo = lo
        self.hi = hi

    def apply(self, tokens: Sequence[str]) -> List[NGram]:
        out: List[NGram] = []
        n_tok = len(tokens)
        for n in range(self.lo, self.hi + 1):
            for i in range(n_tok - n + 1):
                out.append(NGram(tokens[i:i + n]))
        return out

    @classmethod
    def from_saved(cls, config, state):
        return cls(config["lo"], config["hi"])


class NGramsCounts(Transformer):
    """Corpus-level n-gram counts (ref: NGramsCounts; mode "default" sums
    all documents, "noAdd" keeps distinct grams).  Returns a Counter of the
    GLOBAL corpus (merged across ranks)."""

    def __init__(self, mode: str = "default"):
        self.mode = mode

    def apply(self, grams):
        return grams

    def apply_dataset(self, ds: Dataset) -> Counter:
        local = Counter()
        for grams in ds.items():
            if self.mode == "noAdd":
                local.update(set(grams))
            else:
                local.update(grams)
        rt = ds.runtime
        if not rt.is_distributed:
            return local
        import torch.distributed as dist
        gathered = [None] * rt.world_size
        dist.all_gather_object(gathered, local)
        total = Counter()
        for c in gathered:
            total.update(c)
        return total
