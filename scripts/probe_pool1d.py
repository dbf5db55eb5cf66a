import sys, time; sys.path.insert(0, ".")
import torch
from keystone_amd import hip
from keystone_amd.ops.images.sift import _merged_pool_matrix, _triangle_kernel1d

def t(label, fn, iters=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    print(f"{label}: {(time.perf_counter()-t0)/iters*1000:.2f} ms", flush=True)

B8, H, W = 8192 * 8, 256, 256
omap = torch.randn(B8 * H, W, device="cuda")
for (bins, step) in [(6, 6), (8, 6)]:
    import math
    g = math.gcd(step, bins)
    Mw, npos, Q, sy, sj = _merged_pool_matrix(W, bins, step, omap.device, omap.dtype)
    tap = _triangle_kernel1d(bins, omap.device, omap.dtype)
    t(f"GEMM lastdim bin={bins} Q={Q}", lambda: omap @ Mw)
    t(f"kern lastdim bin={bins} Q={Q}", lambda: hip.pool_lastdim(omap, tap, bins // 2, g, Q))
    a = omap @ Mw
    b = hip.pool_lastdim(omap, tap, bins // 2, g, Q)
    print("  lastdim err", (a - b).abs().max().item())
    A1 = a.reshape(B8, H, Q)
    Mh, npy, Qy, syy, sjy = _merged_pool_matrix(H, bins, step, omap.device, omap.dtype)
    t(f"bmm middim bin={bins} Qy={Qy}", lambda: torch.matmul(A1.transpose(1, 2), Mh))
    t(f"kern middim bin={bins} Qy={Qy}", lambda: hip.pool_middim(A1, tap, bins // 2, g, Qy))
    c = torch.matmul(A1.transpose(1, 2), Mh)
    d = hip.pool_middim(A1, tap, bins // 2, g, Qy)
    print("  middim err", (c - d).abs().max().item())
