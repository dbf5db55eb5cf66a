import time, torch
def bench(name, fn, iters=8, warm=2):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t)/iters*1000:.2f} ms", flush=True)

dev = "cuda"
# LCS H-pool shape: [BC, 4nx, H] @ [H, 4ny]
BC, H, nx4, ny4 = 24576, 256, 236, 236
t1 = torch.randn(BC, H, nx4, device=dev)
Mh = torch.randn(H, ny4, device=dev)
bench("lcs bmm broadcast", lambda: torch.matmul(t1.transpose(1, 2), Mh))
def tr_flat():
    a = t1.transpose(1, 2).contiguous()
    return (a.reshape(BC * nx4, H) @ Mh).reshape(BC, nx4, ny4)
bench("lcs transpose+flat", tr_flat)
def mh_first():
    # Mh^T @ t1 as flat: (Mh^T [ny4, H]) @ t1 [BC, H, nx4] -> bmm broadcast A
    return torch.matmul(Mh.T, t1)
bench("lcs MhT@t1 bmm", mh_first)
# SIFT A2 shape: [B8, 4ny, 4nx] = MhT @ A1([B8, H, 4nx])
B8 = 65536; Hs = 256; nxs = 232; nys = 232
A1 = torch.randn(B8, Hs, nxs, device=dev)
Mhs = torch.randn(Hs, nys, device=dev)
bench("sift MhT@A1 bmm", lambda: torch.matmul(Mhs.T, A1))
def sift_tr():
    a = A1.transpose(1, 2).contiguous()
    return (a.reshape(B8 * nxs, Hs) @ Mhs).reshape(B8, nxs, nys)
bench("sift transpose+flat", sift_tr)
