import ctypes, sys, time
import torch
sys.path.insert(0, ".")
from keystone_amd import hip

lib = hip._load()
lib.ks_gemm_ablate_f32.argtypes = [ctypes.c_void_p]*3 + [ctypes.c_int]*4 + [ctypes.c_void_p]
lib.ks_gemm_ablate_f32.restype = ctypes.c_int

n = 8192
A = torch.rand(n, n, device="cuda") * 2 - 1
B = torch.rand(n, n, device="cuda") * 2 - 1
C = torch.empty(n, n, device="cuda")

def run(v):
    rc = lib.ks_gemm_ablate_f32(ctypes.c_void_p(A.data_ptr()), ctypes.c_void_p(B.data_ptr()),
                                ctypes.c_void_p(C.data_ptr()), n, n, n, v,
                                ctypes.c_void_p(torch.cuda.current_stream().cuda_stream))
    assert rc == 0, rc

def bench(v, iters=10):
    for _ in range(3): run(v)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): run(v)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

# interleaved within-probe A/B (guide rule 24)
import statistics
res = {0: [], 1: [], 2: []}
for _ in range(6):
    for v in (0, 1, 2):
        res[v].append(bench(v, 5))
for v in (0, 1, 2):
    t = statistics.median(res[v])
    tf = 2 * n**3 / t / 1e12
    print(f"variant {v}: {t*1e3:.2f} ms = {tf:.1f} TF")
# correctness of full variant
run(0); torch.cuda.synchronize()
ref = A[:512] @ B
err = (C[:512] - ref).abs().max().item()
print("v0 correct:", err < 1e-2 * float(ref.abs().max()), err)
