"""Fisher Vector encoding against a diagonal-covariance GMM.

ref: nodes.images.external.FisherVector (JNI
EncEval ``calcAndGetFVs``) + the Scala twin and OptimizableFisherVector
dispatch.  Improved FV (Perronnin et al.):

  gamma = softmax_k(log w_k + log N(x | mu_k, diag sigma2_k))
  G_mu_k    = (1 / n sqrt(w_k))   sum_i gamma_ik (x_i - mu_k) / sigma_k
  G_sigma_k = (1 / n sqrt(2 w_k)) sum_i gamma_ik [((x_i - mu_k)/sigma_k)^2 - 1]

(power/L2 normalization are separate nodes, as in the reference pipeline:
SignedHellingerMapper -> NormalizeRows).

Identity used by the batched path: with S0 = Gamma^T 1, S1 = Gamma^T X,
S2 = Gamma^T X^2 the sums above close over (S0, S1, S2), so per shard the
whole encoding is two batched GEMMs (hipBLASLt bmm) + elementwise — no
[n, 2KD] intermediate.  The fused one-workgroup-per-image HIP kernel
(posterior + accumulate with GMM params in LDS) is the K3 upgrade and is
dispatched automatically for fp32 CUDA shards.
"""

from __future__ import annotations

import torch

from ...parallel.dataset import Dataset, TensorDataset
from ...workflow.pipeline import Transformer
from ..learning.gmm import GaussianMixtureModel


class FisherVector(Transformer):
    def __init__(self, gmm: GaussianMixtureModel, improved_scaling: bool = True):
        self.gmm = gmm
        self.improved_scaling = improved_scaling

    # ------------------------------------------------------------------
    def _encode(self, X: torch.Tensor) -> torch.Tensor:
        """[B, n, D] (or [n, D]) -> [B, 2*K*D] (or [2*K*D])."""
        squeeze = X.dim() == 2
        if squeeze:
            X = X.unsqueeze(0)
        B, n, D = X.shape
        g = self.gmm
        mu = g.means.to(device=X.device, dtype=X.dtype)        # [K, D]
        var = g.variances.to(device=X.device, dtype=X.dtype)   # [K, D]
        w = g.weights.to(device=X.device, dtype=X.dtype)       # [K]
        K = mu.shape[0]

        import os
        if X.is_cuda and X.dtype == torch.float32 and self.improved_scaling:
            from ... import hip as hip_ops
            hip_ops.gpu_ready()  # extension missing on GPU -> loud failure
            # K3 v2 (fv_mfma.hip): pipelined fused logits-MFMA -> softmax
            # -> moment-MFMA per image; never materializes gamma (10x less
            # HBM traffic), measured 1.6x the batched-GEMM path at K=256
            # (profiles/README.md).
            if (hip_ops.fv_mfma_supported(D, K)
                    and not os.environ.get("KEYSTONE_FV_GEMM")):
                out = hip_ops.fisher_vector_mfma(X.contiguous(), mu, var, w)
                return out.squeeze(0) if squeeze else out
            # first-generation fused kernel: measured slower than the GEMM
            # formulation (LDS-read-issue bound) — kept opt-in
            if (os.environ.get("KEYSTONE_FV_KERNEL")
                    and hip_ops.fv_supported(D, K)):
                out = hip_ops.fisher_vector(X.contiguous(), mu, var, w)
                return out.squeeze(0) if squeeze else out

        # log N(x|k) via the expanded-GEMM form (see gmm.log_prob);
        # X^2 hoisted (shared by the logit GEMM and S2)
        X2 = X * X
        inv2v = 0.5 / var
        a = X2 @ inv2v.T                                        # [B, n, K]
        b = X @ (mu / var).T
        c = (mu * mu * inv2v).sum(1) + 0.5 * var.log().sum(1)
        lp = -a + b - c + w.clamp_min(1e-30).log()
        gamma = torch.softmax(lp, dim=2)                        # [B, n, K]

        S0 = gamma.sum(dim=1)                                   # [B, K]
        S1 = gamma.transpose(1, 2) @ X                          # [B, K, D]
        S2 = gamma.transpose(1, 2) @ X2                         # [B, K, D]

        sig = var.sqrt()
        s0 = S0.unsqueeze(2)
        gmu = (S1 - mu * s0) / sig
        gsig = (S2 - 2 * mu * S1 + (mu * mu) * s0) / var - s0
        if self.improved_scaling:
            gmu = gmu / (n * w.sqrt().unsqueeze(1))
            gsig = gsig / (n * (2 * w).sqrt().unsqueeze(1))
        out = torch.cat([gmu.reshape(B, K * D), gsig.reshape(B, K * D)], dim=1)
        return out.squeeze(0) if squeeze else out

    def apply(self, X: torch.Tensor) -> torch.Tensor:
        return self._encode(X)

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        return self._encode(X)

    def apply_dataset(self, ds: Dataset) -> Dataset:
        if isinstance(ds, TensorDataset) and ds.data.dim() == 3:
            return ds.map_batch(self._encode)
        return ds.map(self._encode)

    # persistence
    def get_state(self):
        return {"means": self.gmm.means, "variances": self.gmm.variances,
                "weights": self.gmm.weights}

    def get_config(self):
        return {"improved_scaling": self.improved_scaling}

    @classmethod
    def from_saved(cls, config, state):
        return cls(GaussianMixtureModel(state["means"], state["variances"],
                                        state["weights"]),
                   config.get("improved_scaling", True))
