"""Dense multi-scale SIFT descriptors.

ref: nodes.images.external.SIFTExtractor -> JNI
``VLFeat.getSIFTs(step, binSize, numScales, img)`` (VLFeat dsift):
Gaussian smoothing per scale, per-pixel gradient magnitude/orientation,
soft orientation binning into 8 bins, triangular (bilinear) spatial
pooling into a 4x4 grid of bins -> 128-dim descriptors on a dense grid,
concatenated over ``num_scales`` scales with bin size doubling.

This implementation is *batched over the whole shard*: every step is a
tensor op ([B, H, W] in, [B, n_desc, 128] out), so the GPU path runs a
fixed chain of MIOpen/rocBLAS/elementwise kernels per shard.  A fused
single-kernel HIP implementation (per-image workgroups, LDS histograms)
is the K1 upgrade; this version doubles as its numerics oracle.
"""

from __future__ import annotations

import math
from typing import Tuple

import torch

from ...parallel.dataset import Dataset, TensorDataset
from ...workflow.pipeline import Transformer

N_ORI = 8
N_SPATIAL = 4  # 4x4 spatial bins -> 128-dim


def _gaussian_kernel1d(sigma: float, device, dtype) -> torch.Tensor:
    radius = max(1, int(math.ceil(3.0 * sigma)))
    x = torch.arange(-radius, radius + 1, device=device, dtype=dtype)
    k = torch.exp(-0.5 * (x / sigma) ** 2)
    return k / k.sum()


_band_cache = {}


def _band_matrix(n: int, k: torch.Tensor, replicate: bool) -> torch.Tensor:
    """[n, n] banded matrix M with out = in @ M equal to a 1-D correlation
    with kernel k (replicate or zero boundary).

    Why: MIOpen lowers tiny separable convs on huge batches to one
    im2col+GEMM PER IMAGE (tens of thousands of ~10 us launches per shard).
    A banded-matrix matmul is ONE hipBLASLt strided-batched GEMM that runs
    at the f32 MFMA rate — ~20x more flops but ~100x fewer launches, a
    large net win on MI355X (measured in profiles/).
    """
    key = (n, k.numel(), replicate, str(k.device), k.dtype,
           round(float(k[0]), 9), round(float(k[k.numel() // 2]), 9))
    M = _band_cache.get(key)
    if M is not None:
        return M
    r = (k.numel() - 1) // 2
    M = torch.zeros(n, n, device=k.device, dtype=k.dtype)
    idx = torch.arange(n, device=k.device)
    for t in range(-r, r + 1):
        j = idx + t
        if replicate:
            M[idx, j.clamp(0, n - 1)] += k[t + r]
        else:
            m = (j >= 0) & (j < n)
            M[idx[m], j[m]] += k[t + r]
    M = M.T.contiguous()  # out[i] = sum_j in[j] M[j, i]
    _band_cache[key] = M
    return M


def _sep_filter(x: torch.Tensor, k: torch.Tensor, replicate: bool
                ) -> torch.Tensor:
    """Separable 2-D correlation on [..., H, W] via two banded GEMMs."""
    H, W = x.shape[-2], x.shape[-1]
    Mw = _band_matrix(W, k, replicate)
    Mh = _band_matrix(H, k, replicate)
    x = x @ Mw                       # along W: batched GEMM
    x = (x.transpose(-1, -2) @ Mh).transpose(-1, -2)
    return x.contiguous()


def _smooth(imgs: torch.Tensor, sigma: float) -> torch.Tensor:
    """Separable Gaussian blur on [B, H, W] (replicate boundary)."""
    if sigma <= 0:
        return imgs
    k = _gaussian_kernel1d(sigma, imgs.device, imgs.dtype)
    return _sep_filter(imgs, k, replicate=True)


def _triangle_kernel1d(bin_size: int, device, dtype) -> torch.Tensor:
    """VLFeat dsift bilinear spatial window: support 2*bin_size-1."""
    x = torch.arange(-(bin_size - 1), bin_size, device=device, dtype=dtype)
    return 1.0 - x.abs() / bin_size


def _sampling_matrix(L: int, bin_size: int, step: int, device, dtype
                     ) -> Tuple[torch.Tensor, int]:
    """[L, N_SPATIAL * npos] matrix folding the triangular spatial pooling
    AND the grid sampling into one banded GEMM: column (j, p) holds the
    triangle window centered at ``bin//2 + p*step + j*bin_size`` (zero
    boundary).  Pooling the dense map then slicing grid points computes
    W-many columns per row; this computes only the 4*npos needed ones."""
    key = ("samp", L, bin_size, step, str(device), dtype)
    hit = _band_cache.get(key)
    if hit is not None:
        return hit
    span = (N_SPATIAL - 1) * bin_size
    anchors = list(range(bin_size // 2, L - span - bin_size // 2, step))
    npos = len(anchors)
    M = torch.zeros(L, N_SPATIAL * npos, device=device, dtype=dtype)
    idx = torch.arange(L, device=device, dtype=dtype)
    for j in range(N_SPATIAL):
        for p, a in enumerate(anchors):
            c = a + j * bin_size
            w = (1.0 - (idx - c).abs() / bin_size).clamp_min(0)
            M[:, j * npos + p] = w
    _band_cache[key] = (M, npos)
    return M, npos


def _merged_pool_matrix(L: int, bin_size: int, step: int, device, dtype):
    """Distinct-cell pooling matrix on the gcd lattice.  The (j, p)
    sampling columns of ``_sampling_matrix`` have centers
    ``bin//2 + p*step + j*bin`` which all lie on the stride-
    ``g = gcd(step, bin)`` lattice: only Q = (npos-1)*step/g + 3*bin/g + 1
    distinct columns exist (neighboring descriptors share pooling cells,
    the classic dsift structure).  Returns (M [L, Q], npos, Q, sy, sj)
    where column q pools at center ``bin//2 + q*g`` and descriptor (p, j)
    reads column ``p*sy + j*sj``."""
    import math as _m
    key = ("mergedg", L, bin_size, step, str(device), dtype)
    hit = _band_cache.get(key)
    if hit is not None:
        return hit
    span = (N_SPATIAL - 1) * bin_size
    anchors = list(range(bin_size // 2, L - span - bin_size // 2, step))
    npos = len(anchors)
    g = _m.gcd(step, bin_size)
    sy, sj = step // g, bin_size // g
    Q = (npos - 1) * sy + (N_SPATIAL - 1) * sj + 1 if npos else 0
    M = torch.zeros(L, max(Q, 1), device=device, dtype=dtype)
    idx = torch.arange(L, device=device, dtype=dtype)
    for q in range(Q):
        c = bin_size // 2 + q * g
        M[:, q] = (1.0 - (idx - c).abs() / bin_size).clamp_min(0)
    _band_cache[key] = (M, npos, Q, sy, sj)
    return M, npos, Q, sy, sj


class SIFTExtractor(Transformer):
    def __init__(self, step: int = 4, bin_size: int = 4, num_scales: int = 2,
                 scale_step: float = math.sqrt(2.0), clamp: float = 0.2,
                 append_xy: bool = False):
        self.step = step
        self.bin_size = bin_size
        self.num_scales = num_scales
        self.scale_step = scale_step
        self.clamp = clamp
        self.append_xy = append_xy

    # ------------------------------------------------------------------
    def _orientation_maps(self, imgs: torch.Tensor, sigma: float
                          ) -> torch.Tensor:
        """[B, H, W] -> [B, 8, H, W] blurred-gradient orientation maps.
        GPU fp32: ONE fused HIP kernel (K1, hip/src/sift.hip); otherwise
        the torch oracle chain."""
        B, H, W = imgs.shape
        dt = imgs.dtype
        if imgs.is_cuda and dt == torch.float32:
            from ... import hip as hip_ops
            if (hip_ops.gpu_ready() and hip_ops.sift_omap_radius(sigma)
                    <= hip_ops.SIFT_OMAP_MAX_RADIUS):
                return hip_ops.sift_omap(imgs.contiguous(), sigma)
        sm = _smooth(imgs, sigma)
        gx = torch.zeros_like(sm)
        gy = torch.zeros_like(sm)
        gx[:, :, 1:-1] = 0.5 * (sm[:, :, 2:] - sm[:, :, :-2])
        gy[:, 1:-1, :] = 0.5 * (sm[:, 2:, :] - sm[:, :-2, :])
        mag = torch.sqrt(gx * gx + gy * gy)
        ang = torch.atan2(gy, gx) % (2 * math.pi)
        t = ang * (N_ORI / (2 * math.pi))
        b0 = torch.floor(t)
        frac = t - b0
        b0 = b0.long() % N_ORI
        b1 = (b0 + 1) % N_ORI
        omap = torch.zeros(B, N_ORI, H, W, device=imgs.device, dtype=dt)
        omap.scatter_add_(1, b0.unsqueeze(1), (mag * (1 - frac)).unsqueeze(1))
        omap.scatter_add_(1, b1.unsqueeze(1), (mag * frac).unsqueeze(1))
        return omap

    def _one_scale(self, imgs: torch.Tensor, bin_size: int, sigma: float,
                   out: torch.Tensor = None) -> torch.Tensor:
        """[B, H, W] -> [B, n, 128] descriptors at one scale."""
        import os
        import time
        dbg = os.environ.get("KEYSTONE_SIFT_DEBUG") and imgs.is_cuda
        _ph = {}

        def tick(name, t0):
            # never sync inside hipGraph capture (invalidates the capture
            # AND poisons the stream for everything after)
            if dbg and not torch.cuda.is_current_stream_capturing():
                torch.cuda.synchronize()
                now = time.perf_counter()
                _ph[name] = _ph.get(name, 0.0) + (now - t0)
                return now
            return t0

        t0 = time.perf_counter()
        B, H, W = imgs.shape
        dt = imgs.dtype
        omap = None   # computed lazily — the fused pooled kernel skips it

        # triangular spatial pooling FUSED with grid sampling: two
        # sampling-banded GEMMs (hipBLASLt) instead of dense pooling of
        # the whole map + slicing (~3.5x fewer flops, no gather pass)
        Mwm, nx_m, Qx, sy_x, sj_x = _merged_pool_matrix(
            W, bin_size, self.step, imgs.device, dt)
        if nx_m and Qx <= N_SPATIAL * nx_m:
            # shared-cell fast path: pool the DISTINCT cells once (gcd
            # lattice), then each descriptor's 4x4 window is an
            # as_strided view of the pooled map; the reshape below
            # performs the gather the old path paid for in contiguous()
            Mw, nx = Mwm, nx_m
            Mh, ny, Qy, sy_y, sj_y = _merged_pool_matrix(
                H, bin_size, self.step, imgs.device, dt)
            if nx == 0 or ny == 0:
                return torch.zeros(B, 0, N_ORI * N_SPATIAL * N_SPATIAL,
                                   device=imgs.device, dtype=dt)
            import os
            A1 = None
            g = math.gcd(self.step, bin_size)
            if (imgs.is_cuda and dt == torch.float32
                    and os.environ.get("KEYSTONE_SIFT_FUSED") == "1"):
                from ... import hip as hip_ops
                if hip_ops.sift_omap_pooled_supported(bin_size):
                    # fused blur+gradient+orientation+pool kernel: never
                    # materializes the [B, 8, H, W] omap and replaces the
                    # banded A1 GEMM.  OPT-IN: numerics-exact (GPU-tested)
                    # but measured SLOWER end-to-end than omap-kernel +
                    # GEMM (78 KB static LDS -> 1 workgroup/CU; extract
                    # 84 -> 107 ms at bench scale).  Needs dynamic LDS +
                    # half-precision orientation planes to win — round-2.
                    A1 = hip_ops.sift_omap_pooled(
                        imgs.contiguous(), sigma, bin_size, g, Qx) \
                        .reshape(B * N_ORI, H, Qx)
            # banded lattice pooling kernel for the W direction when the
            # lattice is dense (small gcd -> Q large): the dense [W, Qx]
            # GEMM spends ~17x the useful flops there (measured: kernel
            # 7.2 vs GEMM 10.4 ms at Q=124; a wash at Q=42, and the
            # middim variant never beat the strided bmm -> GEMM kept)
            use_pool_kernel = False
            if imgs.is_cuda and dt == torch.float32 and Qx >= 64:
                from ... import hip as hip_ops
                use_pool_kernel = hip_ops.gpu_ready()
            if A1 is None:
                omap = self._orientation_maps(imgs, sigma)
                t0 = tick("omap", t0)
                if use_pool_kernel:
                    tap = _triangle_kernel1d(bin_size, imgs.device, dt)
                    A1 = hip_ops.pool_lastdim(
                        omap.reshape(B * N_ORI * H, W), tap,
                        bin_size // 2, g, Qx).reshape(B * N_ORI, H, Qx)
                else:
                    A1 = (omap.reshape(B * N_ORI * H, W) @ Mw) \
                        .reshape(B * N_ORI, H, Qx)
                t0 = tick("a1_gemm", t0)
            # H-pooling as a batched-A @ broadcast-B bmm ([G, m, K] @
            # [K, n] — hipBLASLt strided-batched, the fast form; both the
            # broadcast-A form Mh.T @ [B*8, H, Qx] and a transpose+flat
            # rewrite measured 13 ms on rocBLAS fallbacks / transpose
            # copies).
            P = torch.matmul(A1.transpose(1, 2), Mh)  # [B*8, Qx, Qy]
            t0 = tick("p_gemm", t0)
            # descriptor element order is (o, jx, jy) — OUR convention
            # (PCA/GMM/FV are order-agnostic; fit and serve share it):
            # jy is innermost so the assembly gather reads 4 CONTIGUOUS
            # floats per (o, jx) from P instead of one element per
            # 21k-element orientation plane (o-innermost measured 8.3 ms
            # of gather per step; this order ~3)
            s0, s1, s2 = P.stride()
            src6 = P.as_strided(
                (B, ny, nx, N_ORI, N_SPATIAL, N_SPATIAL),
                (N_ORI * s0, sy_y * s2, sy_x * s1,
                 s0, sj_x * s1, sj_y * s2))
            if out is not None:
                # assemble straight into the caller's [B, n_total, 128]
                # slice (no torch.cat pass across scales): ONE strided
                # copy + the pitch-aware norm kernel
                dst6 = out.as_strided(
                    (B, ny, nx, N_ORI, N_SPATIAL, N_SPATIAL),
                    (out.stride(0), nx * 128, 128, 16, 4, 1))
                dst6.copy_(src6)
                t0 = tick("gather", t0)
                from ... import hip as hip_ops
                hip_ops.sift_norm_slice_(out, self.clamp)
                t0 = tick("norm", t0)
                if dbg:
                    print(f"[sift bin={bin_size}] "
                          + " ".join(f"{k}={v * 1000:.1f}ms"
                                     for k, v in _ph.items()), flush=True)
                return out
            D = src6.reshape(B, ny * nx, N_SPATIAL * N_SPATIAL * N_ORI)
            t0 = tick("gather", t0)
        else:
            Mw, nx = _sampling_matrix(W, bin_size, self.step,
                                      imgs.device, dt)
            Mh, ny = _sampling_matrix(H, bin_size, self.step,
                                      imgs.device, dt)
            if nx == 0 or ny == 0:
                return torch.zeros(B, 0, N_ORI * N_SPATIAL * N_SPATIAL,
                                   device=imgs.device, dtype=dt)
            omap = self._orientation_maps(imgs, sigma)
            A1 = (omap.reshape(B * N_ORI * H, W) @ Mw) \
                .reshape(B * N_ORI, H, -1)             # [B*8, H, 4nx]
            A2 = torch.matmul(Mh.T, A1)                # [B*8, 4ny, 4nx]
            # same (o, jx, jy) element order as the fast path
            D = A2.reshape(B, N_ORI, N_SPATIAL, ny, N_SPATIAL, nx) \
                .permute(0, 3, 5, 1, 4, 2) \
                .reshape(B, ny * nx,
                         N_SPATIAL * N_SPATIAL * N_ORI).contiguous()

        # SIFT normalization: L2 -> clamp 0.2 -> L2 (fused kernel on GPU)
        from ... import hip as hip_ops
        if (D.is_cuda and dt == torch.float32 and D.shape[1] > 0
                and hip_ops.gpu_ready()):
            hip_ops.sift_norm_(D.reshape(-1, D.shape[2]), self.clamp)
        else:
            D = D / D.norm(dim=2, keepdim=True).clamp_min(1e-10)
            D = D.clamp_max(self.clamp)
            D = D / D.norm(dim=2, keepdim=True).clamp_min(1e-10)

        t0 = tick("norm", t0)
        if dbg:
            print(f"[sift bin={bin_size}] "
                  + " ".join(f"{k}={v * 1000:.1f}ms" for k, v in _ph.items()),
                  flush=True)
        if self.append_xy:
            span = (N_SPATIAL - 1) * bin_size
            ys = torch.arange(bin_size // 2, H - span - bin_size // 2,
                              self.step, device=imgs.device)
            xs = torch.arange(bin_size // 2, W - span - bin_size // 2,
                              self.step, device=imgs.device)
            yy = (ys.to(dt) / H - 0.5).repeat_interleave(len(xs))
            xx = (xs.to(dt) / W - 0.5).repeat(len(ys))
            pos = torch.stack([xx, yy], dim=1).unsqueeze(0).expand(B, -1, -1)
            D = torch.cat([D, pos], dim=2)
        return D

    def apply_batch(self, imgs: torch.Tensor) -> torch.Tensor:
        """[B, H, W] -> [B, n_total, 128(+2)] over all scales."""
        if imgs.dim() == 2:
            return self.apply_batch(imgs.unsqueeze(0)).squeeze(0)
        scales = []
        for s in range(self.num_scales):
            bs = max(1, int(round(self.bin_size * (self.scale_step ** s))))
            scales.append((bs, bs / 3.0))  # VLFeat magnif=3 convention

        # GPU fast assembly: when every scale takes the merged lattice
        # path, write each scale's descriptors straight into ONE
        # preallocated [B, n_total, 128] buffer (skips the torch.cat)
        if (imgs.is_cuda and imgs.dtype == torch.float32
                and not self.append_xy):
            from ... import hip as hip_ops
            if hip_ops.gpu_ready():
                B, H, W = imgs.shape
                counts = []
                ok = True
                for bs, _sg in scales:
                    Mwm, nx, Qx, _, _ = _merged_pool_matrix(
                        W, bs, self.step, imgs.device, imgs.dtype)
                    _, ny, _, _, _ = _merged_pool_matrix(
                        H, bs, self.step, imgs.device, imgs.dtype)
                    if not (nx and ny and Qx <= N_SPATIAL * nx):
                        ok = False
                        break
                    counts.append(ny * nx)
                if ok:
                    Dtot = torch.empty(B, sum(counts),
                                       N_SPATIAL * N_SPATIAL * N_ORI,
                                       device=imgs.device, dtype=imgs.dtype)
                    off = 0
                    for (bs, sg), n_s in zip(scales, counts):
                        self._one_scale(imgs, bs, sg,
                                        out=Dtot[:, off:off + n_s])
                        off += n_s
                    return Dtot

        outs = [self._one_scale(imgs, bs, sg) for bs, sg in scales]
        return torch.cat(outs, dim=1)

    def apply(self, img: torch.Tensor) -> torch.Tensor:
        return self.apply_batch(img.unsqueeze(0)).squeeze(0)

    def apply_dataset(self, ds: Dataset) -> Dataset:
        if isinstance(ds, TensorDataset):
            return ds.map_batch(self.apply_batch)
        return ds.map(self.apply)
