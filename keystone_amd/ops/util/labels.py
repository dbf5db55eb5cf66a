"""Label encoding and classification heads.

ref: nodes.util.ClassLabelIndicatorsFromIntLabels (label ->
+-1 one-hot vector), ClassLabelIndicatorsFromStringLabels, MaxClassifier
(argmax), TopKClassifier.
"""

from __future__ import annotations

from typing import Dict, Sequence

import torch

from ...workflow.pipeline import BatchTransformer, Transformer


class ClassLabelIndicatorsFromIntLabels(BatchTransformer):
    """int label -> length-k vector of -1 with +1 at the label position
    (KeystoneML's +-1 encoding feeds the least-squares "classifier")."""

    def __init__(self, num_classes: int):
        self.num_classes = num_classes

    def apply_batch(self, y: torch.Tensor) -> torch.Tensor:
        y = y.long().reshape(-1)
        out = torch.full((y.shape[0], self.num_classes), -1.0, device=y.device)
        out.scatter_(1, y.unsqueeze(1), 1.0)
        return out


class ClassLabelIndicatorsFromStringLabels(Transformer):
    """string label -> +-1 indicator vector via a fitted vocabulary
    (ref: ClassLabelIndicatorsFromStringLabels)."""

    def __init__(self, classes: Sequence[str]):
        self.classes = list(classes)
        self.index: Dict[str, int] = {c: i for i, c in enumerate(self.classes)}

    def apply(self, label: str) -> torch.Tensor:
        out = torch.full((len(self.classes),), -1.0)
        out[self.index[label]] = 1.0
        return out

    def get_config(self):
        return {"classes": self.classes}

    @classmethod
    def from_saved(cls, config, state):
        return cls(config["classes"])


class MaxClassifier(BatchTransformer):
    """argmax over the score vector (ref: MaxClassifier)."""

    def apply_batch(self, scores: torch.Tensor) -> torch.Tensor:
        return scores.argmax(dim=-1)


class TopKClassifier(BatchTransformer):
    """indices of the top-k scores, best first (ref: TopKClassifier)."""

    def __init__(self, k: int):
        self.k = k

    def apply_batch(self, scores: torch.Tensor) -> torch.Tensor:
        return scores.topk(self.k, dim=-1).indices
