"""MFMA FV kernel vs batched-GEMM FV at bench shapes."""
import sys, time
import torch
sys.path.insert(0, ".")
from keystone_amd import hip
from keystone_amd.ops.images.fisher_vector import FisherVector
from keystone_amd.ops.learning.gmm import GaussianMixtureModel

def bench(fn, iters=10, warmup=3):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

torch.manual_seed(0)
for (B, n, D, K) in [(2048, 2900, 64, 256), (8192, 2900, 64, 256),
                     (2048, 2900, 64, 64)]:
    X = torch.randn(B, n, D, device="cuda")
    mu = torch.randn(K, D, device="cuda")
    var = torch.rand(K, D, device="cuda") + 0.5
    w = torch.softmax(torch.randn(K, device="cuda"), 0)
    fv = FisherVector(GaussianMixtureModel(mu, var, w))
    t_mfma = bench(lambda: hip.fisher_vector_mfma(X, mu, var, w))
    import os
    os.environ["KEYSTONE_FV_GEMM"] = "1"
    t_gemm = bench(lambda: fv._encode(X), iters=5)
    del os.environ["KEYSTONE_FV_GEMM"]
    print(f"B{B} n{n} D{D} K{K}: mfma {t_mfma*1e3:.1f} ms, gemm {t_gemm*1e3:.1f} ms, "
          f"speedup {t_gemm/t_mfma:.2f}x")
    del X
    torch.cuda.empty_cache()
