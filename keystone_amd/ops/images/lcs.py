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
    """Distinct-cell box-pooling matrix on the gcd lattice (same trick as
    sift._merged_pool_matrix): subpatch (j, yi) starts at
    ``yi*step + j*s`` — all multiples of ``g = gcd(step, s)`` — so only
    Q = (ny-1)*step/g + 3*s/g + 1 distinct box columns exist.  Returns
    (M [L, Q], ny, Q, sy, sj): subpatch (yi, j) reads column
    ``yi*sy + j*sj``."""
    import math as _m
    key = (L, s, step, str(device), dtype)
    hit = _samp_cache.get(key)
    if hit is not None:
        return hit
    Lm = L - s + 1                      # dense-map length (stride-1 pool)
    span = (N_SUB - 1) * s
    ny = len(range(0, Lm - span, step))
    g = _m.gcd(step, s)
    sy, sj = step // g, s // g
    Q = (ny - 1) * sy + (N_SUB - 1) * sj + 1 if ny else 0
    M = torch.zeros(L, max(Q, 1), device=device, dtype=dtype)
    for q in range(Q):
        st = q * g
        M[st:st + s, q] = 1.0 / s
    _samp_cache[key] = (M, ny, Q, sy, sj)
    return M, ny, Q, sy, sj


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
        Mh, ny, Qy, sy_y, sj_y = _lcs_samp_matrix(
            H, s, self.step, imgs.device, imgs.dtype)
        Mw, nx, Qx, sy_x, sj_x = _lcs_samp_matrix(
            W, s, self.step, imgs.device, imgs.dtype)
        if ny == 0 or nx == 0:
            return torch.zeros(B, 0, 2 * C * N_SUB * N_SUB,
                               device=imgs.device, dtype=imgs.dtype)
        X = imgs.reshape(B * C, H, W)

        def pool(z):
            # W-direction as ONE flat plain GEMM; H-direction batched;
            # output stays in the bmm's contiguous [BC, Qx, Qy] layout
            # (arr() below absorbs the axis order into its strided view)
            t = (z.reshape(B * C * H, W) @ Mw).reshape(B * C, H, -1)
            return torch.matmul(t.transpose(1, 2), Mh)

        m1 = pool(X)
        m2 = pool(X * X)
        std = (m2 - m1 * m1).clamp_min(0).sqrt()

        def arr(m):  # [BC, Qx, Qy] -> [B, ny, nx, j, i, C] lattice view
            P = m.reshape(B, C, Qx, Qy)
            p0, p1, p2, p3 = P.stride()
            return P.as_strided(
                (B, ny, nx, N_SUB, N_SUB, C),
                (p0, sy_y * p3, sy_x * p2, sj_y * p3, sj_x * p2, p1))

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
