import sys, time; sys.path.insert(0, ".")
import torch
import keystone_amd as ka

def t(label, fn, iters=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    print(f"{label}: {(time.perf_counter()-t0)/iters*1000:.2f} ms", flush=True)

X = torch.randn(100_000, 128, device="cuda")
n = X.shape[0]
t("sum+mean fp64", lambda: X.sum(0, dtype=torch.float64))
X64c = X.to(torch.float64)
t("X.to(f64)", lambda: X.to(torch.float64))
mean = X64c.mean(0)
t("center+cast back", lambda: (X.to(torch.float64) - mean).to(X.dtype))
Xc = (X.to(torch.float64) - mean).to(X.dtype)
t("gram fp64", lambda: (lambda z: z.T @ z)(Xc.to(torch.float64)))
G = Xc.to(torch.float64).T @ Xc.to(torch.float64)
t("eigh fp64 128", lambda: torch.linalg.eigh(G))
t("eigh fp32 128", lambda: torch.linalg.eigh(G.float()))
from keystone_amd.ops.learning.pca import DistributedPCAEstimator
t("full fit", lambda: DistributedPCAEstimator(64).fit(ka.from_rows(X)))
