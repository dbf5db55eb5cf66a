"""Image op tests vs naive references (ref: SURVEY.md §4 — Convolver vs
naive conv, Pooler vs loop reference, small synthetic images)."""

import math

import pytest
import torch
import torch.nn.functional as F

import keystone_amd as ka
from keystone_amd.ops.images import (CenterCornerPatcher, Convolver,
                                     DaisyExtractor, FisherVector,
                                     GrayScaler, ImageVectorizer,
                                     LCSExtractor, PixelScaler, Pooler,
                                     RandomPatcher, SIFTExtractor,
                                     SymmetricRectifier, Windower)
from keystone_amd.ops.learning.gmm import GaussianMixtureModel

torch.manual_seed(0)


def test_grayscaler():
    x = torch.rand(3, 8, 8)
    g = GrayScaler().apply(x)
    ref = 0.299 * x[0] + 0.587 * x[1] + 0.114 * x[2]
    assert torch.allclose(g, ref, atol=1e-6)
    assert GrayScaler().apply(torch.rand(8, 8)).shape == (8, 8)


def test_pixel_scaler_and_vectorizer():
    x = (torch.rand(4, 5, 5) * 255).to(torch.uint8)
    y = PixelScaler().apply_batch(x)
    assert y.max() <= 1.0
    v = ImageVectorizer().apply(torch.rand(5, 5))
    assert v.shape == (25,)


def test_windower_matches_manual():
    x = torch.arange(36, dtype=torch.float32).reshape(6, 6)
    p = Windower(stride=2, size=3).apply(x)
    assert p.shape == (4, 9)
    assert torch.equal(p[0], x[0:3, 0:3].reshape(-1))
    assert torch.equal(p[1], x[0:3, 2:5].reshape(-1))


def test_pooler_sum_and_max_vs_loop():
    x = torch.rand(2, 3, 8, 8)
    out = Pooler(4, 4, "sum").apply_batch(x)
    ref = torch.zeros(2, 3, 2, 2)
    for i in range(2):
        for j in range(2):
            ref[:, :, i, j] = x[:, :, 4 * i:4 * i + 4, 4 * j:4 * j + 4].sum((2, 3))
    assert torch.allclose(out, ref, atol=1e-5)
    outm = Pooler(4, 4, "max").apply_batch(x)
    refm = torch.zeros_like(ref)
    for i in range(2):
        for j in range(2):
            refm[:, :, i, j] = x[:, :, 4 * i:4 * i + 4, 4 * j:4 * j + 4].amax((2, 3))
    assert torch.equal(outm, refm)


def test_symmetric_rectifier():
    x = torch.randn(4, 6)
    out = SymmetricRectifier(0.1).apply_batch(x)
    assert out.shape == (4, 12)
    assert torch.allclose(out[:, :6], (x - 0.1).clamp_min(0))
    assert torch.allclose(out[:, 6:], (-x - 0.1).clamp_min(0))


def test_convolver_modes_agree():
    x = torch.rand(2, 1, 16, 16)
    filt = torch.rand(5, 1, 4, 4)
    a = Convolver(filt, "im2col").apply_batch(x)
    b = Convolver(filt, "conv").apply_batch(x)
    c = Convolver(filt, "fft").apply_batch(x)
    ref = F.conv2d(x, filt)
    assert torch.allclose(a, ref, atol=1e-4)
    assert torch.allclose(b, ref, atol=1e-5)
    assert torch.allclose(c, ref, atol=1e-3)


def test_patchers():
    x = torch.rand(12, 12)
    crops = CenterCornerPatcher(8, flips=True).apply(x)
    assert len(crops) == 10
    assert all(c.shape == (8, 8) for c in crops)
    rp = RandomPatcher(3, 5).apply(x)
    assert len(rp) == 3 and rp[0].shape == (5, 5)


def test_sift_shapes_and_norm():
    imgs = torch.rand(2, 40, 40)
    sift = SIFTExtractor(step=4, bin_size=4, num_scales=2)
    D = sift.apply_batch(imgs)
    assert D.dim() == 3 and D.shape[0] == 2 and D.shape[2] == 128
    assert D.shape[1] > 10
    norms = D.norm(dim=2)
    # descriptors are ~unit (clamped-renormalized); flat regions may be 0
    nz = norms[norms > 1e-6]
    assert torch.allclose(nz, torch.ones_like(nz), atol=1e-4)
    assert float(D.max()) <= 0.2 / 0.2 + 1e-5  # clamp at 0.2 then renorm <= 1


def test_sift_oracle_single_descriptor():
    """Slow per-pixel oracle of the 4x4x8 soft-binned descriptor at one
    grid point vs the batched implementation (no smoothing, 1 scale)."""
    torch.manual_seed(1)
    H = W = 24
    img = torch.rand(H, W)
    bs, step = 4, 4
    sift = SIFTExtractor(step=step, bin_size=bs, num_scales=1)
    # replicate: sigma = bs/3 smoothing INSIDE the op; use the same here
    from keystone_amd.ops.images.sift import _smooth, N_ORI
    sm = _smooth(img.unsqueeze(0), bs / 3.0).squeeze(0)
    gx = torch.zeros_like(sm)
    gy = torch.zeros_like(sm)
    gx[:, 1:-1] = 0.5 * (sm[:, 2:] - sm[:, :-2])
    gy[1:-1, :] = 0.5 * (sm[2:, :] - sm[:-2, :])
    mag = (gx * gx + gy * gy).sqrt()
    ang = torch.atan2(gy, gx) % (2 * math.pi)

    D = sift.apply_batch(img.unsqueeze(0)).squeeze(0)
    # first grid point: anchor y0 = x0 = bin_size//2 (bin centers at
    # y0 + j*bs), spatial triangular weight (1-|dy|/bs)(1-|dx|/bs)
    y0 = x0 = bs // 2
    raw = torch.zeros(4, 4, N_ORI, dtype=torch.float64)
    for py in range(H):
        for px in range(W):
            m = float(mag[py, px])
            if m == 0:
                continue
            t = float(ang[py, px]) * N_ORI / (2 * math.pi)
            b0 = int(t) % N_ORI
            frac = t - int(t)
            for j in range(4):
                wy = 1 - abs(py - (y0 + j * bs)) / bs
                if wy <= 0:
                    continue
                for i in range(4):
                    wx = 1 - abs(px - (x0 + i * bs)) / bs
                    if wx <= 0:
                        continue
                    raw[j, i, b0] += m * (1 - frac) * wy * wx
                    raw[j, i, (b0 + 1) % N_ORI] += m * frac * wy * wx
    # the extractor's element order is (o, jx, jy) — see sift.py
    v = raw.permute(2, 1, 0).reshape(-1)
    v = v / v.norm().clamp_min(1e-10)
    v = v.clamp_max(0.2)
    v = v / v.norm().clamp_min(1e-10)
    assert torch.allclose(D[0].to(torch.float64), v, atol=1e-4)


def test_lcs_oracle():
    img = torch.rand(1, 30, 30)
    lcs = LCSExtractor(step=5, subpatch_size=4)
    D = lcs.apply_batch(img.unsqueeze(0)).squeeze(0)
    assert D.shape[1] == 32
    # first descriptor, first subpatch stats
    patch = img[0, 0:4, 0:4]
    assert abs(float(D[0, 0]) - float(patch.mean())) < 1e-5
    assert abs(float(D[0, 16]) - float(patch.std(unbiased=False))) < 1e-4


def test_daisy_shapes():
    imgs = torch.rand(2, 40, 40)
    d = DaisyExtractor(step=6)
    D = d.apply_batch(imgs)
    assert D.shape[0] == 2 and D.shape[2] == (1 + 2 * 8) * 8  # 136


def test_fisher_vector_oracle():
    torch.manual_seed(2)
    K, Dd, n = 4, 6, 50
    means = torch.randn(K, Dd)
    variances = torch.rand(K, Dd) + 0.5
    weights = torch.softmax(torch.randn(K), dim=0)
    gmm = GaussianMixtureModel(means, variances, weights)
    fv = FisherVector(gmm)
    X = torch.randn(n, Dd)
    out = fv.apply(X)
    assert out.shape == (2 * K * Dd,)

    # naive per-descriptor oracle
    import numpy as np
    lp = gmm.log_prob(X)
    gamma = torch.softmax(lp, dim=1)
    sig = variances.sqrt()
    gmu = torch.zeros(K, Dd, dtype=torch.float64)
    gsig = torch.zeros(K, Dd, dtype=torch.float64)
    for i in range(n):
        for k in range(K):
            u = (X[i] - means[k]) / sig[k]
            gmu[k] += float(gamma[i, k]) * u.double()
            gsig[k] += float(gamma[i, k]) * (u.double() ** 2 - 1)
    for k in range(K):
        gmu[k] /= n * math.sqrt(float(weights[k]))
        gsig[k] /= n * math.sqrt(2 * float(weights[k]))
    ref = torch.cat([gmu.reshape(-1), gsig.reshape(-1)])
    assert torch.allclose(out.to(torch.float64), ref, atol=1e-4)


def test_fisher_vector_batched_matches_single():
    K, Dd = 3, 5
    gmm = GaussianMixtureModel(torch.randn(K, Dd), torch.rand(K, Dd) + 0.5,
                               torch.softmax(torch.randn(K), 0))
    fv = FisherVector(gmm)
    X = torch.randn(4, 20, Dd)
    batched = fv.apply_batch(X)
    for b in range(4):
        single = fv.apply(X[b])
        assert torch.allclose(batched[b], single, atol=1e-5)


def test_lcs_matches_dense_avgpool_reference():
    # gcd-lattice shared-cell pooling == dense stride-1 avg_pool + sampling
    import torch.nn.functional as F
    from keystone_amd.ops.images.lcs import LCSExtractor, N_SUB
    torch.manual_seed(3)
    imgs = torch.rand(2, 3, 64, 60)
    e = LCSExtractor(step=4, subpatch_size=6)
    D = e.apply_batch(imgs)
    s, step = 6, 4
    B, C, H, W = imgs.shape
    mean = F.avg_pool2d(imgs, s, stride=1)
    mean2 = F.avg_pool2d(imgs * imgs, s, stride=1)
    std = (mean2 - mean * mean).clamp_min(0).sqrt()
    span = (N_SUB - 1) * s
    ys = torch.arange(0, mean.shape[2] - span, step)
    xs = torch.arange(0, mean.shape[3] - span, step)
    parts = []
    for m in (mean, std):
        for j in range(N_SUB):
            for i in range(N_SUB):
                parts.append(m[:, :, ys + j * s][:, :, :, xs + i * s])
    Dref = torch.stack(parts, dim=2).permute(0, 3, 4, 2, 1) \
        .reshape(B, len(ys) * len(xs), -1)
    assert D.shape == Dref.shape
    assert (D - Dref).abs().max().item() < 1e-5


def test_sift_merged_pooling_matches_sampling_matrix():
    # the gcd-lattice fast path == the general (j, p)-column formulation
    import keystone_amd.ops.images.sift as m
    torch.manual_seed(4)
    B, H, W = 2, 96, 80
    imgs = torch.rand(B, H, W)
    e = m.SIFTExtractor(step=4, bin_size=4, num_scales=1)
    NS, NO = m.N_SPATIAL, m.N_ORI
    for bs in (4, 6):  # gcd 4 and gcd 2 cases
        omap = e._orientation_maps(imgs, 1.2)
        dt = imgs.dtype
        Mw, nx = m._sampling_matrix(W, bs, 4, imgs.device, dt)
        Mh, ny = m._sampling_matrix(H, bs, 4, imgs.device, dt)
        A1 = (omap.reshape(B * NO * H, W) @ Mw).reshape(B * NO, H, -1)
        A2 = torch.matmul(Mh.T, A1)
        Dref = A2.reshape(B, NO, NS, ny, NS, nx).permute(0, 3, 5, 1, 4, 2) \
            .reshape(B, ny * nx, NS * NS * NO).contiguous()
        Mwm, nxm, Qx, syx, sjx = m._merged_pool_matrix(W, bs, 4,
                                                       imgs.device, dt)
        Mhm, nym, Qy, syy, sjy = m._merged_pool_matrix(H, bs, 4,
                                                       imgs.device, dt)
        assert (nxm, nym) == (nx, ny)
        A1q = (omap.reshape(B * NO * H, W) @ Mwm).reshape(B * NO, H, Qx)
        P = torch.matmul(Mhm.T, A1q).reshape(B, NO, Qy, Qx)
        s0, s1, s2, s3 = P.stride()
        Dfast = P.as_strided((B, ny, nx, NO, NS, NS),
                             (s0, syy * s2, syx * s3, s1,
                              sjx * s3, sjy * s2)) \
            .reshape(B, ny * nx, NS * NS * NO)
        assert (Dref - Dfast).abs().max().item() < 1e-5


def test_fast_windower_matches_windower():
    from keystone_amd.ops.images.basic import FastWindower, Windower
    torch.manual_seed(4)
    for shape in [(2, 3, 17, 23), (3, 16, 16)]:
        X = torch.randn(*shape)
        for stride, size in [(2, 5), (3, 4)]:
            slow = Windower(stride, size).apply_batch(X)
            fast = FastWindower(stride, size).apply_batch(X)
            assert torch.equal(slow, fast), (shape, stride, size)
    x = torch.randn(3, 12, 14)
    assert torch.equal(Windower(2, 4).apply(x), FastWindower(2, 4).apply(x))


def test_random_image_transformer_flip_and_crop():
    from keystone_amd.ops.images import RandomImageTransformer
    x = torch.arange(24.0).reshape(4, 6)
    always = RandomImageTransformer(flip_prob=1.0, seed=1)
    assert torch.equal(always.apply(x), x.flip(-1))
    never = RandomImageTransformer(flip_prob=0.0, crop_size=3, seed=2)
    out = never.apply(x)
    assert out.shape[-2:] == (3, 3)


def test_padded_fft_matches_manual():
    from keystone_amd.ops.stats import PaddedFFT
    torch.manual_seed(13)
    X = torch.randn(5, 6)                  # pads to 8
    out = PaddedFFT().apply_batch(X)
    ref = torch.fft.rfft(X, n=8, dim=-1).abs()[..., 1:]
    assert torch.allclose(out, ref, atol=1e-6)
    assert out.shape == (5, 4)             # rfft(8) -> 5 bins, DC dropped


def test_linear_rectifier_semantics():
    from keystone_amd.ops.stats import LinearRectifier
    x = torch.tensor([[-1.0, 0.2, 2.0]])
    out = LinearRectifier(max_val=0.0, alpha=0.5).apply_batch(x)
    assert torch.equal(out, torch.tensor([[0.0, 0.0, 1.5]]))


def test_convolver_matches_naive_loop_oracle():
    """All three modes vs a quadruple-loop correlation oracle — cross-mode
    agreement alone could hide a shared convention error (flip vs no
    flip, offset)."""
    torch.manual_seed(4)
    B, C, H, W, Fn, hh, ww = 2, 3, 9, 8, 4, 3, 3
    X = torch.randn(B, C, H, W)
    filt = torch.randn(Fn, C, hh, ww)
    Ho, Wo = H - hh + 1, W - ww + 1
    ref = torch.zeros(B, Fn, Ho, Wo, dtype=torch.float64)
    Xd, fd = X.to(torch.float64), filt.to(torch.float64)
    for b in range(B):
        for f in range(Fn):
            for i in range(Ho):
                for j in range(Wo):
                    ref[b, f, i, j] = (Xd[b, :, i:i + hh, j:j + ww]
                                       * fd[f]).sum()
    for mode in ("im2col", "conv", "fft"):
        out = Convolver(filt, mode=mode).apply_batch(X)
        assert torch.allclose(out.to(torch.float64), ref, atol=1e-4), mode


def test_sift_three_scales_shapes_and_norm():
    """Multi-scale beyond the bench default: descriptor counts follow the
    per-scale grids (bin x sqrt(2) per scale, VLFeat convention) and
    normalization holds."""
    import math as _m
    imgs = torch.rand(2, 64, 64)
    s3 = SIFTExtractor(step=4, bin_size=4, num_scales=3)
    D = s3.apply_batch(imgs)
    # per-scale anchor count along one axis for L=64:
    # anchors = range(bin//2, L - 3*bin - bin//2, step)
    def npos(L, b, st):
        return len(range(b // 2, L - 3 * b - b // 2, st))
    bins = [max(1, int(round(4 * (_m.sqrt(2.0) ** s)))) for s in range(3)]
    assert bins == [4, 6, 8]
    expect = sum(npos(64, b, 4) ** 2 for b in bins)
    assert D.shape == (2, expect, 128)
    nz = D.norm(dim=2)
    nz = nz[nz > 1e-6]
    assert torch.allclose(nz, torch.ones_like(nz), atol=1e-4)
