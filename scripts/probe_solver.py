"""Standalone bench-scale weighted-solver probe (GPU)."""
import os, time, torch
import keystone_amd as ka
from keystone_amd.ops.learning.block_ls import BlockWeightedLeastSquaresEstimator

torch.manual_seed(0)
n, d, K = 8192, 65536, 1000
X = torch.randn(n, d, device="cuda")
cls = torch.randint(0, K, (n,), device="cuda")
Y = torch.full((n, K), -1.0, device="cuda")
Y.scatter_(1, cls.unsqueeze(1), 1.0)
ds, ls = ka.from_rows(X), ka.from_rows(Y)
est = BlockWeightedLeastSquaresEstimator(4096, 1, 1e-5, 0.5)
est._fit(ds, ls)  # warm
os.environ["KEYSTONE_SOLVER_DEBUG"] = "1"
torch.cuda.synchronize(); t = time.perf_counter()
est._fit(ds, ls)
torch.cuda.synchronize()
print(f"solver fit: {time.perf_counter()-t:.3f} s", flush=True)
