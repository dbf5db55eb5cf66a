"""Daisy dense descriptors.

ref: nodes.images.DaisyExtractor — Gaussian-smoothed
orientation maps sampled at a center plus rings of points (Tola et al.):
descriptor = concat of L2-normalized 8-bin orientation histograms at the
center (finest smoothing) and at ``n_points`` per ring (coarser smoothing
per ring) -> (1 + rings*points) * 8 dims (default 200).
"""

from __future__ import annotations

import math

import torch

from ...parallel.dataset import Dataset, TensorDataset
from ...workflow.pipeline import Transformer
from .sift import N_ORI, _smooth


class DaisyExtractor(Transformer):
    def __init__(self, step: int = 4, radii=(5, 10), n_points: int = 8,
                 sigmas=(2.5, 5.0, 7.5), eps: float = 1e-10):
        assert len(sigmas) == len(radii) + 1
        self.step = step
        self.radii = tuple(radii)
        self.n_points = n_points
        self.sigmas = tuple(sigmas)
        self.eps = eps

    def _orientation_maps(self, imgs: torch.Tensor) -> torch.Tensor:
        B, H, W = imgs.shape
        gx = torch.zeros_like(imgs)
        gy = torch.zeros_like(imgs)
        gx[:, :, 1:-1] = 0.5 * (imgs[:, :, 2:] - imgs[:, :, :-2])
        gy[:, 1:-1, :] = 0.5 * (imgs[:, 2:, :] - imgs[:, :-2, :])
        mag = torch.sqrt(gx * gx + gy * gy)
        ang = torch.atan2(gy, gx) % (2 * math.pi)
        t = ang * (N_ORI / (2 * math.pi))
        b0 = torch.floor(t)
        frac = t - b0
        b0 = b0.long() % N_ORI
        b1 = (b0 + 1) % N_ORI
        omap = torch.zeros(B, N_ORI, H, W, device=imgs.device, dtype=imgs.dtype)
        omap.scatter_add_(1, b0.unsqueeze(1), (mag * (1 - frac)).unsqueeze(1))
        omap.scatter_add_(1, b1.unsqueeze(1), (mag * frac).unsqueeze(1))
        return omap

    def apply_batch(self, imgs: torch.Tensor) -> torch.Tensor:
        if imgs.dim() == 2:
            return self.apply_batch(imgs.unsqueeze(0)).squeeze(0)
        B, H, W = imgs.shape
        omap = self._orientation_maps(imgs)
        # smooth orientation maps at each ring's sigma
        smoothed = []
        for sg in self.sigmas:
            sm = _smooth(omap.reshape(B * N_ORI, H, W), sg)
            smoothed.append(sm.reshape(B, N_ORI, H, W))

        margin = int(max(self.radii)) + 1
        ys = torch.arange(margin, H - margin, self.step, device=imgs.device)
        xs = torch.arange(margin, W - margin, self.step, device=imgs.device)
        if len(ys) == 0 or len(xs) == 0:
            dim = (1 + len(self.radii) * self.n_points) * N_ORI
            return torch.zeros(B, 0, dim, device=imgs.device, dtype=imgs.dtype)
        gy = ys.reshape(-1, 1).expand(-1, len(xs))
        gx = xs.reshape(1, -1).expand(len(ys), -1)

        def grab(level: int, dy: int, dx: int) -> torch.Tensor:
            h = smoothed[level][:, :, gy + dy, gx + dx]     # [B, 8, ny, nx]
            h = h / h.norm(dim=1, keepdim=True).clamp_min(self.eps)
            return h

        hists = [grab(0, 0, 0)]
        for ri, radius in enumerate(self.radii):
            for p in range(self.n_points):
                th = 2 * math.pi * p / self.n_points
                dy = int(round(radius * math.sin(th)))
                dx = int(round(radius * math.cos(th)))
                hists.append(grab(ri + 1, dy, dx))
        D = torch.stack(hists, dim=2)  # [B, 8, P, ny, nx]
        return D.permute(0, 3, 4, 2, 1).reshape(B, len(ys) * len(xs), -1)

    def apply(self, img: torch.Tensor) -> torch.Tensor:
        return self.apply_batch(img.unsqueeze(0)).squeeze(0)

    def apply_dataset(self, ds: Dataset) -> Dataset:
        if isinstance(ds, TensorDataset):
            return ds.map_batch(self.apply_batch)
        return ds.map(self.apply)
