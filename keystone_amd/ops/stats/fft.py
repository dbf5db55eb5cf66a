"""Padded FFT featurizer.

ref: nodes.stats.PaddedFFT — zero-pad each row to the next
power of two, real FFT, keep magnitudes (MNIST RandomFFT featurizer).

MI355X path: torch.fft on ROCm dispatches to rocFFT (batched R2C) — this is
a library-GEMM-class op, no hand kernel needed.
"""

from __future__ import annotations

import torch

from ...workflow.pipeline import BatchTransformer


class PaddedFFT(BatchTransformer):
    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        d = X.shape[-1]
        n = 1 << (d - 1).bit_length()  # next power of two
        spec = torch.fft.rfft(X, n=n, dim=-1)
        # keep magnitudes, drop the DC bin (matches keeping d "interesting"
        # coefficients; KeystoneML returned the magnitude vector)
        return spec.abs()[..., 1:]
