"""Random cosine features (Rahimi–Recht random kitchen sinks).

ref: nodes.stats.CosineRandomFeatures — the TIMIT/MNIST
featurizer: ``z = cos(x W + b)`` (optionally scaled), with ``W ~ N(0, gamma)``
(d x D) and ``b ~ U[0, 2 pi]``.

MI355X path: the GEMM + cos is ONE fused HIP kernel (MFMA f32 GEMM with a
cos epilogue — kernel ``gemm_cos_f32`` in keystone_amd/hip) so the D-wide
intermediate ``xW + b`` never round-trips through HBM.  CPU path: torch.
"""

from __future__ import annotations

import math

import torch

from ...workflow.pipeline import BatchTransformer


class CosineRandomFeatures(BatchTransformer):
    def __init__(self, input_dim: int, num_features: int, gamma: float,
                 use_scale: bool = True, seed: int = 0,
                 dtype: torch.dtype = torch.float32):
        g = torch.Generator().manual_seed(seed)
        self.input_dim = input_dim
        self.num_features = num_features
        self.gamma = gamma
        self.seed = seed
        # W ~ N(0, gamma): KeystoneML draws Gaussian entries scaled by gamma.
        self.W = torch.randn(input_dim, num_features, generator=g,
                             dtype=torch.float32).to(dtype) * math.sqrt(gamma)
        self.b = (torch.rand(num_features, generator=g, dtype=torch.float32)
                  .to(dtype) * (2 * math.pi))
        self.scale = math.sqrt(2.0 / num_features) if use_scale else 1.0

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        W = self.W.to(device=X.device, dtype=X.dtype)
        b = self.b.to(device=X.device, dtype=X.dtype)
        self.W, self.b = W, b  # keep resident on the compute device
        if X.is_cuda and X.dtype == torch.float32:
            from ... import hip as hip_ops
            if hip_ops.gpu_ready():
                return hip_ops.gemm_cos(X, W, b, self.scale)
        return torch.cos(X @ W + b) * self.scale

    def get_config(self):
        return {"input_dim": self.input_dim, "num_features": self.num_features,
                "gamma": self.gamma, "seed": self.seed, "scale": self.scale}

    def get_state(self):
        return {"W": self.W, "b": self.b}

    @classmethod
    def from_saved(cls, config, state):
        obj = cls.__new__(cls)
        obj.__dict__.update(config)
        obj.W = state["W"]
        obj.b = state["b"]
        return obj
