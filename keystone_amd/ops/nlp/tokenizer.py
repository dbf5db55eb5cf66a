"""Text preprocessing transformers.

ref: nodes.nlp.Tokenizer (regex split), LowerCase, Trim.
Tokenization stays CPU-side (strings never touch the GPU); the output of
the NLP chain becomes sparse CSR tensors via CommonSparseFeatures.
"""

from __future__ import annotations

import re
from typing import List

from ...workflow.pipeline import Transformer


class Tokenizer(Transformer):
    """Split on a regex (default: non-word chars), dropping empties."""

    def __init__(self, pattern: str = r"[\s\W]+"):
        self.pattern = pattern
        self._rx = re.compile(pattern)

    def apply(self, text: str) -> List[str]:
        return [t for t in self._rx.split(text) if t]

    @classmethod
    def from_saved(cls, config, state):
        return cls(config.get("pattern", r"[\s\W]+"))


class LowerCase(Transformer):
    def apply(self, x):
        if isinstance(x, str):
            return x.lower()
        return [t.lower() for t in x]


class Trim(Transformer):
    def apply(self, x):
        if isinstance(x, str):
            return x.strip()
        return [t.strip() for t in x if t.strip()]
