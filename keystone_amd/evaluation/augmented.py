"""Vote over augmented examples.

ref: evaluation.AugmentedExamplesEvaluator — ImageNet test
pipelines score several crops per image; predictions sharing an image id
are averaged before the argmax.
"""

from __future__ import annotations

import torch

from .classification import MulticlassClassifierEvaluator, MulticlassMetrics, \
    _as_label_tensor


class AugmentedExamplesEvaluator:
    def __init__(self, num_classes: int):
        self.num_classes = num_classes

    def evaluate(self, scores, image_ids, actuals) -> MulticlassMetrics:
        S = _as_label_tensor(scores).to(torch.float64)
        ids = _as_label_tensor(image_ids).long().reshape(-1)
        act = _as_label_tensor(actuals).long().reshape(-1)
        uniq, inv = ids.unique(return_inverse=True)
        agg = torch.zeros(uniq.shape[0], S.shape[1], dtype=S.dtype,
                          device=S.device)
        agg.index_add_(0, inv, S)
        # one occurrence's label per image (augmentations of the same
        # image share a label, so which occurrence wins is immaterial)
        lab = torch.zeros(uniq.shape[0], dtype=torch.long, device=S.device)
        lab.scatter_(0, inv, act)
        pred = agg.argmax(1)
        return MulticlassClassifierEvaluator(self.num_classes).evaluate(pred, lab)
