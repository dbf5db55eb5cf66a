"""Local Color Statistics descriptors.

ref: nodes.images.LCSExtractor — at each dense grid point,
the mean and standard deviation of each channel over a 4x4 grid of
subpatches -> 4*4*2*C dims (96 for RGB, 32 for gray).  Pure tensor ops
(avg_pool sums + slicing), batched over the shard.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from ...parallel.dataset import Dataset, TensorDataset
from ...workflow.pipeline import Transformer

N_SUB = 4  # 4x4 subpatch grid


class LCSExtractor(Transformer):
    def __init__(self, step: int = 4, subpatch_size: int = 6, eps: float = 1e-10):
        self.step = step
        self.subpatch_size = subpatch_size
        self.eps = eps

    def apply_batch(self, imgs: torch.Tensor) -> torch.Tensor:
        """[B, C?, H, W] -> [B, n, 2*C*16]"""
        if imgs.dim() == 2:
            return self.apply_batch(imgs.unsqueeze(0)).squeeze(0)
        if imgs.dim() == 3:
            imgs = imgs.unsqueeze(1)
        B, C, H, W = imgs.shape
        s = self.subpatch_size
        # per-subpatch mean and E[x^2] maps (stride-1, sampled later)
        mean = F.avg_pool2d(imgs, s, stride=1)
        mean2 = F.avg_pool2d(imgs * imgs, s, stride=1)
        std = (mean2 - mean * mean).clamp_min(0).sqrt()
        Hm, Wm = mean.shape[2], mean.shape[3]

        span = (N_SUB - 1) * s
        ys = torch.arange(0, Hm - span, self.step, device=imgs.device)
        xs = torch.arange(0, Wm - span, self.step, device=imgs.device)
        if len(ys) == 0 or len(xs) == 0:
            return torch.zeros(B, 0, 2 * C * N_SUB * N_SUB,
                               device=imgs.device, dtype=imgs.dtype)
        parts = []
        for m in (mean, std):
            for j in range(N_SUB):
                for i in range(N_SUB):
                    parts.append(m[:, :, ys + j * s][:, :, :, xs + i * s])
        D = torch.stack(parts, dim=2)  # [B, C, 32, ny, nx]
        return D.permute(0, 3, 4, 2, 1).reshape(B, len(ys) * len(xs), -1)

    def apply(self, img: torch.Tensor) -> torch.Tensor:
        return self.apply_batch(img.unsqueeze(0)).squeeze(0)

    def apply_dataset(self, ds: Dataset) -> Dataset:
        if isinstance(ds, TensorDataset):
            return ds.map_batch(self.apply_batch)
        return ds.map(self.apply)
