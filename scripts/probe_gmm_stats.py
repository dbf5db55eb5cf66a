import sys; sys.path.insert(0, ".")
"""A/B the fused STATS E-step vs eager at bench scale (GPU probe)."""
import os, time, torch
from keystone_amd import hip
from keystone_amd.ops.learning.gmm import GaussianMixtureModelEstimator
import keystone_amd as ka

torch.manual_seed(0)
n, D, K = 100_000, 64, 256
X = torch.randn(n, D, device="cuda")

# raw E-step A/B
import math
mu = torch.randn(K, D, device="cuda"); var = torch.rand(K, D, device="cuda") + .5
w = torch.softmax(torch.randn(K, device="cuda"), 0)
def eager():
    inv2v = 0.5 / var
    cc = ((mu*mu*inv2v).sum(1) + 0.5*var.log().sum(1)
          + 0.5*D*math.log(2*math.pi) - w.clamp_min(1e-30).log())
    Xext = torch.cat([X*X, X], 1)
    P = torch.cat([-inv2v, mu/var], 1)
    lp = torch.addmm(-cc, Xext, P.T)
    lse = torch.logsumexp(lp, 1)
    g = torch.exp(lp - lse.unsqueeze(1))
    return g.sum(0, dtype=torch.float64), (g.T @ X).double(), (g.T @ (X*X)).double(), lse.sum(dtype=torch.float64)
def fused():
    return hip.gmm_stats_mfma(X, mu, var, w)
for name, fn in [("eager", eager), ("fused", fused)]:
    for _ in range(3): fn()
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(10): fn()
    torch.cuda.synchronize()
    print(f"estep {name}: {(time.perf_counter()-t)/10*1000:.2f} ms", flush=True)

# full fit A/B (KEYSTONE_GMM_DEBUG prints kmeans_init / em_loop split)
os.environ["KEYSTONE_GMM_DEBUG"] = "1"
ds = ka.from_rows(X)
for tag, env in [("fused", {}), ("eager", {"KEYSTONE_GMM_EAGER": "1"})]:
    for k, v in env.items(): os.environ[k] = v
    est = GaussianMixtureModelEstimator(K, max_iterations=25, seed=0)
    est._fit(ds)  # warm
    torch.cuda.synchronize(); t = time.perf_counter()
    m = est._fit(ds)
    torch.cuda.synchronize()
    print(f"gmm fit ({tag}): {time.perf_counter()-t:.3f} s  w[0]={m.weights[0].item():.5f}", flush=True)
    for k in env: del os.environ[k]
