"""Per-document term frequency with a weighting function.

ref: nodes.misc.TermFrequency — token/n-gram list ->
{term: weight(count)} (e.g. sqrt TF for the Amazon pipeline).
"""

from __future__ import annotations

import math
from collections import Counter
from typing import Dict, Sequence

from ...workflow.pipeline import Transformer

_WEIGHTS = {
    "identity": lambda x: x,
    "sqrt": math.sqrt,
    "log": lambda x: math.log(1.0 + x),
    "binary": lambda x: 1.0,
}


class TermFrequency(Transformer):
    def __init__(self, weighting: str = "identity"):
        self.weighting = weighting

    def apply(self, terms: Sequence) -> Dict:
        fn = _WEIGHTS[self.weighting]
        return {t: float(fn(c)) for t, c in Counter(terms).items()}

    @classmethod
    def from_saved(cls, config, state):
        return cls(config.get("weighting", "identity"))
