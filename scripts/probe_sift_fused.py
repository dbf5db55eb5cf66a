import os, time, torch
from keystone_amd.ops.images.sift import SIFTExtractor
torch.manual_seed(0)
imgs = torch.rand(8192, 256, 256, device="cuda")
e = SIFTExtractor()  # step 4; bench uses bin 6, 2 scales
e.bin_size, e.num_scales = 6, 2
for tag, env in [("gemm", {}), ("fused", {"KEYSTONE_SIFT_FUSED": "1"})]:
    for k, v in env.items(): os.environ[k] = v
    e.apply_batch(imgs); e.apply_batch(imgs)
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(3): D = e.apply_batch(imgs)
    torch.cuda.synchronize()
    print(f"sift extract ({tag}): {(time.perf_counter()-t)/3*1000:.1f} ms", flush=True)
    for k in env: del os.environ[k]
