"""Local Color Statistics descriptors.

ref: nodes.images.LCSExtractor — at each dense grid point,
the mean and standard deviation of each channel over a 4x4 grid of
subpatches -> 4*4*2*C dims (96 for RGB, 32 for gray).

MI355X path: the subpatch box means are folded into TWO sampled banded
GEMMs per moment (one along H, one along W) that compute ONLY the grid
points the 4x4 subpatch layout touches — torch's stride-1 ``avg_pool2d``
ran 21 ms per call on [8192, 3, 256, 256] and computed a dense map of
which the descriptor grid reads ~1/4, and the 32-slice gather loop that
followed cost another ~15 ms of index kernels (profiles/
pipeline_kernel_stats.csv).  The GEMM form runs at the MFMA rate and the
(j, i) subpatch axes fall out of one reshape.
"""

from __future__ import annotations

import torch

from ...parallel.dataset import Dataset, TensorDataset
from ...workflow.pipeline import Transformer

N_SUB = 4  # 4x4 subpatch grid

_samp_cache: dict = {}


def _lcs_samp_matrix(L: int, s: int, step: int, device, dtype):
    """[L, N_SUB * ny] matrix: column (j, yi) averages rows
    ``ys[yi] + j*s .. +s-1`` (the box window of subpatch j at grid row yi).
    Column order is j-major so the output reshapes to [..., N_SUB, ny]."""
    key = (L, s, step, str(device), dtype)
    hit = _samp_cache.get(key)
    if hit is not None:
        return hit
    Lm = L - s + 1                      # dense-map length (stride-1 pool)
    span = (N_SUB - 1) * s
    ys = list(range(0, Lm - span, step))
    ny = len(ys)
    M = torch.zeros(L, N_SUB * ny, device=device, dtype=dtype)
    for j in range(N_SUB):
        for yi, y0 in enumerate(ys):
            st = y0 + j * s
            M[st:st + s, j * ny + yi] = 1.0 / s
    _samp_cache[key] = (M, ny)
    return M, ny


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
        Mh, ny = _lcs_samp_matrix(H, s, self.step, imgs.device, imgs.dtype)
        Mw, nx = _lcs_samp_matrix(W, s, self.step, imgs.device, imgs.dtype)
        if ny == 0 or nx == 0:
            return torch.zeros(B, 0, 2 * C * N_SUB * N_SUB,
                               device=imgs.device, dtype=imgs.dtype)
        X = imgs.reshape(B * C, H, W)

        def pool(z):
            # W-direction as ONE flat plain GEMM; H-direction batched;
            # output stays in the bmm's contiguous [BC, 4*nx, 4*ny] layout
            # (arr() below absorbs the axis order into its view)
            t = (z.reshape(B * C * H, W) @ Mw).reshape(B * C, H, -1)
            return torch.matmul(t.transpose(1, 2), Mh)

        m1 = pool(X)
        m2 = pool(X * X)
        std = (m2 - m1 * m1).clamp_min(0).sqrt()

        def arr(m):  # [BC, 4*nx, 4*ny] -> [B, ny, nx, j, i, C] view
            return m.reshape(B, C, N_SUB, nx, N_SUB, ny) \
                    .permute(0, 5, 3, 4, 2, 1)

        # one contiguous write builds the final layout
        # [(stat*16 + j*4 + i) * C + c] — identical to the reference's
        # mean/std-major ordering
        D = torch.stack([arr(m1), arr(std)], dim=3)
        return D.reshape(B, ny * nx, 2 * N_SUB * N_SUB * C)

    def apply(self, img: torch.Tensor) -> torch.Tensor:
        return self.apply_batch(img.unsqueeze(0)).squeeze(0)

    def apply_dataset(self, ds: Dataset) -> Dataset:
        if isinstance(ds, TensorDataset):
            return ds.map_batch(self.apply_batch)
        return ds.map(self.apply)
