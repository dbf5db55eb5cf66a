import sys, time
import torch
sys.path.insert(0, ".")
from keystone_amd import hip
B, n, D, K = 2048, 2900, 64, 256
X = torch.randn(B, n, D, device="cuda")
mu = torch.randn(K, D, device="cuda")
var = torch.rand(K, D, device="cuda") + 0.5
w = torch.softmax(torch.randn(K, device="cuda"), 0)
for _ in range(3):
    hip.fisher_vector_mfma(X, mu, var, w)
torch.cuda.synchronize()
print("done")
