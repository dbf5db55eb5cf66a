"""Probe: GMM E-step variants head-to-head on the bench shape (GPU)."""
import math
import sys
import time

import torch

sys.path.insert(0, ".")
from keystone_amd import hip
from keystone_amd.ops.learning.gmm import GaussianMixtureModel


def bench(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


torch.manual_seed(0)
n, D, K = 100000, 64, 64
X = torch.randn(n, D, device="cuda")
X2 = X * X
mu = torch.randn(K, D, device="cuda")
var = torch.rand(K, D, device="cuda") + 0.5
w = torch.softmax(torch.randn(K, device="cuda"), 0)
model = GaussianMixtureModel(mu, var, w)


def estep_old():
    lp = model.log_prob(X)
    lse = torch.logsumexp(lp, dim=1)
    gamma = torch.exp(lp - lse.unsqueeze(1))
    Nk = gamma.sum(0, dtype=torch.float64)
    Sx = (gamma.T @ X).to(torch.float64)
    Sxx = (gamma.T @ (X * X)).to(torch.float64)
    return Nk


def estep_addmm():
    inv2v = 0.5 / var
    cc = (mu * mu * inv2v).sum(1) + 0.5 * var.log().sum(1) \
        + 0.5 * D * math.log(2 * math.pi) - w.clamp_min(1e-30).log()
    lp = torch.addmm(-cc, X, (mu / var).T)
    lp = lp.addmm_(X2, inv2v.T, alpha=-1.0)
    lse = torch.logsumexp(lp, dim=1)
    gamma = torch.exp(lp - lse.unsqueeze(1))
    Nk = gamma.sum(0, dtype=torch.float64)
    Sx = (gamma.T @ X).to(torch.float64)
    Sxx = (gamma.T @ X2).to(torch.float64)
    return Nk


def estep_hip():
    gamma, ll = hip.gmm_posteriors(X, mu, var, w, want_ll=True)
    Nk = gamma.sum(0, dtype=torch.float64)
    Sx = (gamma.T @ X).to(torch.float64)
    Sxx = (gamma.T @ X2).to(torch.float64)
    return Nk


print(f"estep old   : {bench(estep_old)*1e3:.2f} ms")
print(f"estep addmm : {bench(estep_addmm)*1e3:.2f} ms")
print(f"estep hip   : {bench(estep_hip)*1e3:.2f} ms")

# where does the time go inside addmm variant?
inv2v = 0.5 / var
bT = (mu / var).T.contiguous()
cc = torch.zeros(K, device="cuda")
print(f"  gemm1 (X@bT)      : {bench(lambda: torch.addmm(-cc, X, bT))*1e3:.2f} ms")
lp0 = torch.addmm(-cc, X, bT)
print(f"  logsumexp         : {bench(lambda: torch.logsumexp(lp0, dim=1))*1e3:.2f} ms")
lse0 = torch.logsumexp(lp0, dim=1)
print(f"  exp-sub           : {bench(lambda: torch.exp(lp0 - lse0.unsqueeze(1)))*1e3:.2f} ms")
g0 = torch.exp(lp0 - lse0.unsqueeze(1))
print(f"  gamma.T@X         : {bench(lambda: g0.T @ X)*1e3:.2f} ms")
print(f"  sum fp64          : {bench(lambda: g0.sum(0, dtype=torch.float64))*1e3:.2f} ms")
