import time, torch
from keystone_amd.ops.images.sift import SIFTExtractor
import keystone_amd.ops.images.sift as m
from keystone_amd import hip as hip_ops

torch.manual_seed(0)
B, H, W = 8192, 256, 256
imgs = torch.rand(B, H, W, device="cuda")
e = SIFTExtractor()

def t(name, fn, iters=3):
    for _ in range(2): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): r = fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t0)/iters*1000:.2f} ms", flush=True)
    return r

t("apply_batch total", lambda: e.apply_batch(imgs), 2)
omap = t("omap", lambda: e._orientation_maps(imgs, 1.2))
dt = imgs.dtype
NS, NO = m.N_SPATIAL, m.N_ORI
Mw, nx, Qx, syx, sjx = m._merged_pool_matrix(W, 4, 4, imgs.device, dt)
Mh, ny, Qy, syy, sjy = m._merged_pool_matrix(H, 4, 4, imgs.device, dt)
A1 = t("A1 flat gemm", lambda: (omap.reshape(B*NO*H, W) @ Mw).reshape(B*NO, H, Qx))
P = t("P bmm", lambda: torch.matmul(Mh.T, A1).reshape(B, NO, Qy, Qx))
def gather():
    s0, s1, s2, s3 = P.stride()
    return P.as_strided((B, ny, nx, NS, NS, NO),
                        (s0, syy*s2, syx*s3, sjy*s2, sjx*s3, s1)) \
            .reshape(B, ny*nx, NS*NS*NO)
D = t("D gather", gather)
t("norm", lambda: hip_ops.sift_norm_(D.reshape(-1, 128), 0.2))
t("one_scale", lambda: e._one_scale(imgs, 4, 1.2), 2)
