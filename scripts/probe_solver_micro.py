"""Micro A/B: pre_chol split + wb_gemm variants + small-solve variants."""
import time, torch

def bench(name, fn, iters=10, warm=3):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t)/iters*1000:.2f} ms", flush=True)

torch.manual_seed(0)
dev = "cuda"
# --- pre_chol split (16 x 4096^2 SPD)
A = torch.randn(16, 4096, 4096, device=dev)
Bb = A @ A.transpose(1, 2) / 4096 + 10 * torch.eye(4096, device=dev)
bench("batched cholesky", lambda: torch.linalg.cholesky(Bb))
Lb = torch.linalg.cholesky(Bb)
bench("batched cholesky_inverse", lambda: torch.cholesky_inverse(Lb))
eye = torch.eye(4096, device=dev).expand(16, 4096, 4096)
def inv_via_trsm():
    Linv = torch.linalg.solve_triangular(Lb, eye, upper=False)
    return Linv.transpose(1, 2) @ Linv
bench("inverse via trtri+gemm", inv_via_trsm)

# --- wb_gemm variants ([8000,4096] @ [4096,4096])
U = torch.randn(8000, 4096, device=dev)
Binv = torch.cholesky_inverse(torch.linalg.cholesky(Bb[0]))
out = torch.empty(8000, 4096, device=dev)
bench("NN gemm", lambda: U @ Binv)
bench("NN gemm out=", lambda: torch.matmul(U, Binv, out=out))
outT = torch.empty(4096, 8000, device=dev)
bench("TN gemm (Binv@U.T)", lambda: torch.matmul(Binv, U.T, out=outT))

# --- small batched solve variants ([1000, 8, 8] SPD)
S = torch.randn(1000, 8, 8, device=dev)
S = S @ S.transpose(1, 2) + 4 * torch.eye(8, device=dev)
t_ = torch.randn(1000, 8, 1, device=dev)
bench("LU solve 8x8x1000", lambda: torch.linalg.solve(S, t_))
def chol_solve():
    return torch.cholesky_solve(t_, torch.linalg.cholesky(S))
bench("chol solve 8x8x1000", chol_solve)
def cpu_solve():
    z = torch.linalg.solve(S.cpu(), t_.cpu())
    return z.to(dev)
bench("cpu roundtrip solve", cpu_solve)
# S bmm variant check: [1000,8,4096]@[1000,4096,8]
Ub = torch.randn(1000, 8, 4096, device=dev)
UBb = torch.randn(1000, 8, 4096, device=dev)
bench("S bmm 8x4096x8", lambda: Ub @ UBb.transpose(1, 2))
