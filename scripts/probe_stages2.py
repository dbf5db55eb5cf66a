import sys, time; sys.path.insert(0, ".")
import torch
import keystone_amd as ka
from keystone_amd.ops.learning.block_ls import _chol_linv
from keystone_amd.ops.learning.gmm import GaussianMixtureModelEstimator
from keystone_amd.ops.learning.pca import DistributedPCAEstimator

def t(label, fn, iters=3):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    print(f"{label}: {(time.perf_counter()-t0)/iters*1000:.2f} ms", flush=True)

# chol base sweep
nbq, db = 16, 4096
A = torch.randn(nbq, db, db//2, device="cuda")
B = A @ A.transpose(1, 2) + db * torch.eye(db, device="cuda")
for base in (256, 512, 1024, 2048, 4096):
    t(f"chol_linv base={base}", lambda b=base: _chol_linv(B, b))

# GMM fit breakdown (bench vocab shape)
X = torch.randn(100_000, 64, device="cuda")
ds = ka.from_rows(X)
import os
os.environ["KEYSTONE_GMM_DEBUG"] = "1"
g = GaussianMixtureModelEstimator(256, max_iterations=20, seed=0)
torch.cuda.synchronize(); t0 = time.perf_counter()
g.fit(ds)
torch.cuda.synchronize()
print(f"gmm fit total: {(time.perf_counter()-t0)*1000:.1f} ms", flush=True)
del os.environ["KEYSTONE_GMM_DEBUG"]

# PCA fit + project (bench shapes)
Xs = torch.randn(100_000, 128, device="cuda")
t("pca fit [100k,128]->64", lambda: DistributedPCAEstimator(64).fit(ka.from_rows(Xs)))
p = DistributedPCAEstimator(64).fit(ka.from_rows(Xs))
D = torch.randn(8192 * 1369, 128, device="cuda")
t("pca project [11.2M,128]", lambda: p.apply_batch(D), 2)
