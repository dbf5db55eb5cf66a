"""Verify the assumed v_mfma_f32_32x32x16_f16 A/B/D fragment layout."""
import ctypes
import sys

import torch

sys.path.insert(0, ".")
import keystone_amd.hip as hip

lib = hip._load()
lib.ks_probe_mfma16.argtypes = [ctypes.c_void_p] * 3 + [ctypes.c_void_p]
lib.ks_probe_mfma16.restype = ctypes.c_int

torch.manual_seed(0)
# asymmetric integer matrices, exactly representable in f16
A = torch.randint(0, 64, (32, 16)).to(torch.float16).cuda()
B = torch.randint(0, 64, (16, 32)).to(torch.float16).cuda()
D = torch.zeros(32, 32, dtype=torch.float32, device="cuda")
rc = lib.ks_probe_mfma16(A.data_ptr(), B.data_ptr(), D.data_ptr(),
                         hip._stream())
torch.cuda.synchronize()
assert rc == 0, rc
ref = (A.float() @ B.float())
ok = torch.equal(D, ref)
print("layout exact match:", ok)
if not ok:
    diff = (D - ref).abs()
    print("max err", diff.max().item())
    print("D[0,:8]", D[0, :8].tolist())
    print("ref[0,:8]", ref[0, :8].tolist())
    # try to diagnose transposes
    print("D==ref.T:", torch.equal(D, ref.T))
