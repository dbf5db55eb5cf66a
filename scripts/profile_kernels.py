"""Minimal kernel exercise for rocprofv3 profiling (keeps traces small).
Runs each hand-written HIP kernel a few times at benchmark-like shapes."""
import sys

import torch

sys.path.insert(0, ".")
from keystone_amd import hip

torch.manual_seed(0)
dev = "cuda"

# MFMA f32 GEMM, 256^2-tile path (4096^3, uniform random)
A = torch.rand(4096, 4096, device=dev) * 2 - 1
B = torch.rand(4096, 4096, device=dev) * 2 - 1
for _ in range(3):
    hip.gemm_fused(A, B, None, hip.EP_NONE)

# fused cos epilogue (random-features shape)
X = torch.randn(16384, 512, device=dev)
W = torch.randn(512, 4096, device=dev) * 0.1
b = torch.rand(4096, device=dev)
for _ in range(3):
    hip.gemm_fused(X, W, b, hip.EP_COS, scale=0.1)   # fully-fused MFMA path
    hip.bias_cos_((X @ W).contiguous(), b, 0.1)      # hybrid path epilogue

# K1 SIFT orientation maps (bench image shape)
imgs = torch.rand(64, 256, 256, device=dev)
for _ in range(3):
    om = hip.sift_omap(imgs, sigma=2.0)

# SIFT descriptor normalization
D = torch.rand(64 * 1500, 128, device=dev)
for _ in range(3):
    hip.sift_norm_(D, 0.2)

# fused signed-sqrt + L2 row normalization (FV post-processing shape)
F = torch.randn(2048, 16384, device=dev)
for _ in range(3):
    hip.hellinger_l2(F)

# K2 GMM posteriors
Xg = torch.randn(100000, 64, device=dev)
mu = torch.randn(64, 64, device=dev)
var = torch.rand(64, 64, device=dev) + 0.5
w = torch.softmax(torch.randn(64, device=dev), 0)
for _ in range(3):
    hip.gmm_posteriors(Xg, mu, var, w, want_ll=True)

# K3 Fisher Vector (fused per-image kernel)
Xd = torch.randn(64, 1500, 64, device=dev)
for _ in range(3):
    hip.fisher_vector(Xd, mu, var, w)

torch.cuda.synchronize()
print("profiled kernel set complete")

# K3 v2: fused MFMA Fisher Vector (the production FV path) + STATS mode
Xf = torch.randn(256, 2900, 64, device=dev)
mu = torch.randn(256, 64, device=dev)
var = torch.rand(256, 64, device=dev) + 0.5
w = torch.softmax(torch.randn(256, device=dev), 0)
for _ in range(3):
    hip.fisher_vector_mfma(Xf, mu, var, w)
Xs = torch.randn(100_000, 64, device=dev)
for _ in range(3):
    hip.gmm_stats_mfma(Xs, mu, var, w)
torch.cuda.synchronize()

# round-2 kernels ------------------------------------------------------
# two-launch GMM EM pieces
Xg = torch.randn(100_000, 64, device=dev)
mu = torch.randn(256, 64, device=dev)
var = torch.rand(256, 64, device=dev) + 0.5
w = torch.softmax(torch.randn(256, device=dev), 0)
a, b2, c2 = hip._gmm_coeffs(mu, var, w)
P = hip._fv_P(a, b2, c2, 256, Xg.device)
st = torch.empty(512, 256 + 2 * 256 * 64 + 1, device=dev)
mu_b, var_b, w_b = mu.clone(), var.clone(), w.clone()
ll_b = torch.zeros(1, device=dev)
for _ in range(3):
    hip.gmm_stats_raw(Xg, P, st)
    pre = st.sum(0, dtype=torch.float64)
    hip.gmm_mstep_(st, pre, 100_000.0, 1e-6, mu_b, var_b, w_b, P, ll_b)

# per-class Woodbury + segsum (solver shapes: db=4096, K=1000, nc~8)
n, db, K = 8192, 4096, 1000
T_all = torch.randn(n, db, device=dev)
Tm = torch.randn(n, K, device=dev)
cls = torch.sort(torch.randint(0, K, (n,), device=dev)).values
counts = torch.bincount(cls, minlength=K)
starts = (counts.cumsum(0) - counts).to(torch.int32)
ncs = counts.to(torch.int32)
cols = torch.arange(K, dtype=torch.int32, device=dev)
LRt = torch.randn(K, db, device=dev)
for _ in range(3):
    hip.woodbury_yz(T_all, LRt, starts, ncs)
    hip.segsum(T_all, Tm, starts, ncs, cols)

# banded lattice pooling (bench scale-2 shape)
from keystone_amd.ops.images.sift import _triangle_kernel1d
om2 = torch.randn(64 * 8 * 256, 256, device=dev)
tap = _triangle_kernel1d(8, om2.device, om2.dtype)
for _ in range(3):
    hip.pool_lastdim(om2, tap, 4, 2, 124)

# pitch-aware descriptor norm
Dt = torch.randn(64, 2900, 128, device=dev)
for _ in range(3):
    hip.sift_norm_slice_(Dt[:, 100:1400], 0.2)
torch.cuda.synchronize()
print("profile exercise done")
