"""Quick GPU probe: HIP kernel perf sanity (run on the MI355X box)."""
import sys
import time

import torch

sys.path.insert(0, ".")
from keystone_amd import hip


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    dev = "cuda"
    torch.manual_seed(0)
    print(torch.cuda.get_device_name(0))

    # f32 MFMA GEMM at 4096^3 (guide's untuned ref: ~122 TF, ceiling 155)
    for n in (2048, 4096, 8192):
        A = torch.rand(n, n, device=dev) * 2 - 1
        B = torch.rand(n, n, device=dev) * 2 - 1
        t = bench(lambda: hip.gemm_fused(A, B, None, hip.EP_NONE))
        tf = 2 * n**3 / t / 1e12
        # rocBLAS comparison
        tr = bench(lambda: A @ B)
        tf_roc = 2 * n**3 / tr / 1e12
        print(f"gemm_fused f32 {n}^3: {t*1e3:.2f} ms = {tf:.1f} TF "
              f"(rocBLAS {tf_roc:.1f} TF)")

    # fused cos GEMM at random-features shape (TIMIT-ish)
    M, K, N = 65536, 512, 4096
    X = torch.randn(M, K, device=dev)
    W = torch.randn(K, N, device=dev) * 0.1
    b = torch.rand(N, device=dev)
    t = bench(lambda: hip.gemm_cos(X, W, b, 0.1))
    import os; os.environ["KEYSTONE_FUSED_GEMM_COS"]="1"
    t_full = bench(lambda: hip.gemm_cos(X, W, b, 0.1)); del os.environ["KEYSTONE_FUSED_GEMM_COS"]
    tf = 2 * M * K * N / t / 1e12
    t_eager = bench(lambda: torch.cos(X @ W + b) * 0.1)
    print(f"gemm_cos {M}x{K}x{N}: hybrid {t*1e3:.2f} ms, fullfused {t_full*1e3:.2f} ms = {tf:.1f} TF "
          f"(eager matmul+cos {t_eager*1e3:.2f} ms, "
          f"speedup {t_eager/t:.2f}x)")

    # fused hellinger+l2 (memory-bound)
    X = torch.randn(4096, 32768, device=dev)
    t = bench(lambda: hip.hellinger_l2(X))
    gbs = 3 * X.numel() * 4 / t / 1e9
    def eager():
        y = torch.sign(X) * torch.sqrt(torch.abs(X))
        return y / y.norm(dim=1, keepdim=True).clamp_min(1e-6)
    t_e = bench(eager)
    print(f"hellinger_l2 {tuple(X.shape)}: {t*1e3:.2f} ms = {gbs:.0f} GB/s "
          f"eff (eager {t_e*1e3:.2f} ms, speedup {t_e/t:.2f}x)")


if __name__ == "__main__":
    main()
