import sys, time; sys.path.insert(0, ".")
import torch
from keystone_amd import hip
from keystone_amd.ops.learning.block_ls import _blocked_trtri

n, D, K = 100_000, 64, 256
X = torch.randn(n, D, device="cuda")
mu = torch.randn(K, D, device="cuda"); var = torch.rand(K, D, device="cuda") + .5
w = torch.softmax(torch.randn(K, device="cuda"), 0)
a, b, c = hip._gmm_coeffs(mu, var, w)
P = hip._fv_P(a, b, c, K, X.device)
nb = 512
width = K + 2 * K * D + 1
out = torch.empty(nb, width, device="cuda")
mu_b, var_b = mu.clone(), var.clone()
w_b = w.clone(); ll_b = torch.zeros(1, device="cuda")

def t(label, fn, iters=20):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    print(f"{label}: {(time.perf_counter()-t0)/iters*1000:.3f} ms", flush=True)

t("estep stats_raw", lambda: hip.gmm_stats_raw(X, P, out))
t("mstep kernel", lambda: hip.gmm_mstep_(out, None, float(n), 1e-6, mu_b, var_b, w_b, P, ll_b))
def old_mstep():
    s = out.sum(0, dtype=torch.float64)
    Nk = s[:K]; Sx = s[K:K+K*D].view(K, D); Sxx = s[K+K*D:K+2*K*D].view(K, D)
    Nk_c = Nk.clamp_min(1e-10).unsqueeze(1)
    m64 = Sx / Nk_c
    v64 = (Sxx / Nk_c - m64 ** 2).clamp_min(1e-6)
    mdl_mu = m64.float(); mdl_var = v64.float(); mdl_w = (Nk / n).float()
    inv2v = 0.5 / mdl_var
    P[:D].copy_(-inv2v.T)
    P[64:64+D].copy_((mdl_mu / mdl_var).T)
    import math
    P[128].copy_(mdl_w.clamp_min(1e-30).log() - (mdl_mu*mdl_mu*inv2v).sum(1)
                 - 0.5*mdl_var.log().sum(1) - 0.5*D*math.log(2*math.pi))
t("old torch mstep chain", old_mstep)

# trtri
nbq, db = 16, 4096
A = torch.randn(nbq, db, db//2, device="cuda")
B = A @ A.transpose(1, 2) + db * torch.eye(db, device="cuda")
L = torch.linalg.cholesky(B)
eye = torch.eye(db, device="cuda")
t("solve_triangular(eye)", lambda: torch.linalg.solve_triangular(L, eye.expand(nbq, db, db), upper=False), 3)
t("blocked_trtri", lambda: _blocked_trtri(L), 3)
I1 = torch.linalg.solve_triangular(L, eye.expand(nbq, db, db), upper=False)
I2 = _blocked_trtri(L)
print("trtri err:", (I1 - I2).abs().max().item())

from keystone_amd.ops.learning.block_ls import _chol_linv
t("chol+trtri current", lambda: _blocked_trtri(torch.linalg.cholesky(B)), 3)
t("schur _chol_linv", lambda: _chol_linv(B), 3)
I3 = _chol_linv(B)
Lref = torch.linalg.cholesky(B)
Iref = torch.linalg.solve_triangular(Lref, eye.expand(nbq, db, db), upper=False)
print("chol_linv err:", ((I3 - Iref).abs().max() / Iref.abs().max()).item())
