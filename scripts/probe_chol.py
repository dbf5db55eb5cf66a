import sys; sys.path.insert(0, ".")
"""Micro-probe: fastest way to get B^{-1} (or an applicable factor) for
nb=16 SPD matrices of size 4096 (the weighted solver pre-pass, 51.5 ms
measured).  Candidates:
  1. cholesky + solve_triangular(eye) + Linv^T@Linv   (current)
  2. cholesky + cholesky_inverse (rocSOLVER potri)
  3. cholesky + keep Linv only (defer the last GEMM to use sites)
  4. torch.linalg.inv (getrf/getri)
"""
import time

import torch

nb, db = 16, 4096
torch.manual_seed(0)
A = torch.randn(nb, db, db // 2, device="cuda")
B = A @ A.transpose(1, 2) + db * torch.eye(db, device="cuda")
eye = torch.eye(db, device="cuda")


def t(label, fn, iters=3):
    for _ in range(1):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print(f"{label}: {(time.perf_counter()-t0)/iters*1000:.1f} ms", flush=True)


t("cholesky only", lambda: torch.linalg.cholesky(B))
L = torch.linalg.cholesky(B)
t("solve_triangular(eye)", lambda: torch.linalg.solve_triangular(
    L, eye.expand(nb, db, db), upper=False))
Linv = torch.linalg.solve_triangular(L, eye.expand(nb, db, db), upper=False)
t("Linv^T @ Linv", lambda: Linv.transpose(1, 2) @ Linv)
t("cholesky_inverse", lambda: torch.cholesky_inverse(L))
t("linalg.inv", lambda: torch.linalg.inv(B))
t("full current path", lambda: (lambda Lb: torch.linalg.solve_triangular(
    Lb, eye.expand(nb, db, db), upper=False))(torch.linalg.cholesky(B)))

# applying the factor: UB = U @ Binv (8192 x 4096 per block) vs two TRSMs
U = torch.randn(nb, 8192, db, device="cuda")
Binv = Linv.transpose(1, 2) @ Linv
t("U @ Binv (bmm)", lambda: U @ Binv, 2)
t("two TRSM apply", lambda: torch.linalg.solve_triangular(
    L.transpose(1, 2), torch.linalg.solve_triangular(
        L, U.transpose(1, 2), upper=False), upper=True), 2)
