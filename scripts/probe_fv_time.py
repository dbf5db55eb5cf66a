import sys; sys.path.insert(0, ".")
import time, torch
from keystone_amd import hip
def run(B, n, D, K, iters=5):
    X = torch.randn(B, n, D, device="cuda")
    mu = torch.randn(K, D, device="cuda")
    var = torch.rand(K, D, device="cuda") + 0.5
    w = torch.softmax(torch.randn(K, device="cuda"), 0)
    for _ in range(2): hip.fisher_vector_mfma(X, mu, var, w)
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(iters): hip.fisher_vector_mfma(X, mu, var, w)
    torch.cuda.synchronize()
    print(f"fv_mfma B={B} n={n} D={D} K={K}: {(time.perf_counter()-t)/iters*1000:.2f} ms", flush=True)
run(2048, 2900, 64, 256)
run(8192, 2900, 64, 256, 3)
run(2048, 2900, 64, 64)
run(2048, 2900, 64, 128)
