"""Augmentation croppers.

ref: nodes.images.CenterCornerPatcher, RandomPatcher,
RandomImageTransformer (ImageNet/VOC train & test augmentation).
"""

from __future__ import annotations

from typing import List

import torch

from ...workflow.pipeline import Transformer


def _crop(x: torch.Tensor, top: int, left: int, size: int) -> torch.Tensor:
    if x.dim() == 2:
        return x[top:top + size, left:left + size]
    return x[:, top:top + size, left:left + size]


class CenterCornerPatcher(Transformer):
    """center + 4 corner crops, optionally horizontally flipped (10 views)
    (ref: CenterCornerPatcher)."""

    def __init__(self, size: int, flips: bool = True):
        self.size = size
        self.flips = flips

    def apply(self, x: torch.Tensor) -> List[torch.Tensor]:
        H, W = x.shape[-2], x.shape[-1]
        s = self.size
        spots = [(0, 0), (0, W - s), (H - s, 0), (H - s, W - s),
                 ((H - s) // 2, (W - s) // 2)]
        out = [_crop(x, t, l, s) for t, l in spots]
        if self.flips:
            out += [p.flip(-1) for p in out]
        return out


class RandomPatcher(Transformer):
    """n random crops per image (ref: RandomPatcher)."""

    def __init__(self, num_patches: int, size: int, seed: int = 0):
        self.num_patches = num_patches
        self.size = size
        self.seed = seed

    def apply(self, x: torch.Tensor) -> List[torch.Tensor]:
        H, W = x.shape[-2], x.shape[-1]
        g = torch.Generator().manual_seed(self.seed)
        out = []
        for _ in range(self.num_patches):
            t = int(torch.randint(0, H - self.size + 1, (1,), generator=g))
            l = int(torch.randint(0, W - self.size + 1, (1,), generator=g))
            out.append(_crop(x, t, l, self.size))
        return out


class RandomImageTransformer(Transformer):
    """random horizontal flip (+ optional random crop) augmentation
    (ref: RandomImageTransformer)."""

    def __init__(self, flip_prob: float = 0.5, crop_size: int = 0, seed: int = 0):
        self.flip_prob = flip_prob
        self.crop_size = crop_size
        self.seed = seed
        self._g = torch.Generator().manual_seed(seed)

    def apply(self, x: torch.Tensor) -> torch.Tensor:
        if float(torch.rand(1, generator=self._g)) < self.flip_prob:
            x = x.flip(-1)
        if self.crop_size:
            H, W = x.shape[-2], x.shape[-1]
            t = int(torch.randint(0, H - self.crop_size + 1, (1,), generator=self._g))
            l = int(torch.randint(0, W - self.crop_size + 1, (1,), generator=self._g))
            x = _crop(x, t, l, self.crop_size)
        return x
