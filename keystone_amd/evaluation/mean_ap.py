"""Mean average precision (VOC-style).

ref: evaluation.MeanAveragePrecisionEvaluator — per-class
average precision over ranked scores (11-point interpolated option), for
multi-label tasks (VOC 2007).
"""

from __future__ import annotations

import torch

from ..parallel.runtime import get_runtime
from .classification import _as_label_tensor


class MeanAveragePrecisionEvaluator:
    def __init__(self, num_classes: int, interpolated_11pt: bool = False):
        self.num_classes = num_classes
        self.interpolated_11pt = interpolated_11pt

    def evaluate(self, scores, actuals) -> dict:
        """scores: [n, K]; actuals: [n, K] binary (multi-label) or [n] int."""
        S = _as_label_tensor(scores).to(torch.float64)
        A = _as_label_tensor(actuals)
        if A.dim() == 1:
            A = torch.nn.functional.one_hot(A.long(), self.num_classes)
        A = A.to(torch.float64)
        rt = get_runtime()
        # gather shards (scores are small: n x K); exact global ranking
        S = rt.all_gather_rows(S.contiguous())
        A = rt.all_gather_rows(A.contiguous())
        aps = []
        for c in range(self.num_classes):
            order = S[:, c].argsort(descending=True)
            rel = A[order, c]
            n_pos = float(rel.sum())
            if n_pos == 0:
                continue
            cum = rel.cumsum(0)
            ranks = torch.arange(1, rel.shape[0] + 1, dtype=torch.float64,
                                 device=rel.device)
            prec = cum / ranks
            if self.interpolated_11pt:
                recall = cum / n_pos
                ap = 0.0
                for t in torch.linspace(0, 1, 11, dtype=torch.float64):
                    mask = recall >= t
                    ap += float(prec[mask].max()) / 11 if bool(mask.any()) else 0.0
            else:
                ap = float((prec * rel).sum() / n_pos)
            aps.append(ap)
        mean_ap = sum(aps) / max(len(aps), 1)
        return {"mAP": mean_ap, "per_class_AP": aps}
