"""Dense patch convolution of image banks.

ref: nodes.images.Convolver — the CIFAR random-patch
pipeline convolves every image with a bank of whitened patches, im2col +
BLAS GEMM in the reference.

MI355X paths:
  * ``mode="im2col"`` — unfold + one hipBLASLt GEMM per shard (the
    reference's formulation, library GEMM);
  * ``mode="fft"``   — rocFFT R2C batched convolution (BASELINE.json's
    "FFT convolution" capability) — wins for large filters;
  * ``mode="conv"``  — MIOpen direct conv2d.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from ...workflow.pipeline import BatchTransformer


class Convolver(BatchTransformer):
    def __init__(self, filters: torch.Tensor, mode: str = "im2col",
                 whitener: Optional[torch.Tensor] = None,
                 filter_mean: Optional[torch.Tensor] = None):
        """filters: [F, C, hh, ww] (or [F, hh, ww] for gray).
        Optional ZCA ``whitener``/``filter_mean`` are folded into the
        filter bank (the reference pre-whitens patches before the GEMM)."""
        if filters.dim() == 3:
            filters = filters.unsqueeze(1)
        self.mode = mode
        self.filters = filters
        if whitener is not None:
            Fb = filters.reshape(filters.shape[0], -1)
            mu = filter_mean if filter_mean is not None else 0.0
            Fb = (Fb - mu) @ whitener.T if isinstance(mu, torch.Tensor) else Fb @ whitener.T
            self.filters = Fb.reshape_as(filters)

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        """[B, C?, H, W] -> [B, F, H-hh+1, W-ww+1] (valid conv, correlation
        semantics like the reference's patch matching)."""
        if X.dim() == 3:
            X = X.unsqueeze(1)
        W = self.filters.to(device=X.device, dtype=X.dtype)
        if self.mode == "conv":
            return F.conv2d(X, W)
        if self.mode == "fft":
            return self._fft_conv(X, W)
        # im2col + GEMM (hipBLASLt via matmul)
        Fn, C, hh, ww = W.shape
        cols = F.unfold(X, (hh, ww))                      # [B, C*hh*ww, L]
        out = W.reshape(Fn, -1) @ cols                    # [B, F, L] (bmm)
        Ho, Wo = X.shape[2] - hh + 1, X.shape[3] - ww + 1
        return out.reshape(X.shape[0], Fn, Ho, Wo)

    @staticmethod
    def _fft_conv(X: torch.Tensor, W: torch.Tensor) -> torch.Tensor:
        B, C, H, Wd = X.shape
        Fn, _C, hh, ww = W.shape
        fh, fw = H, Wd
        Xf = torch.fft.rfft2(X, s=(fh, fw))               # [B, C, fh, fw//2+1]
        # correlation = conv with flipped kernel; flip then conv == correlate
        Wf = torch.fft.rfft2(W.flip(-1).flip(-2), s=(fh, fw))
        Yf = torch.einsum("bchw,fchw->bfhw", Xf, Wf)
        Y = torch.fft.irfft2(Yf, s=(fh, fw))
        return Y[:, :, hh - 1: H, ww - 1: Wd].contiguous()

    def apply(self, x):
        return self.apply_batch(x.unsqueeze(0)).squeeze(0)

    def get_state(self):
        return {"filters": self.filters}

    def get_config(self):
        return {"mode": self.mode}

    @classmethod
    def from_saved(cls, config, state):
        return cls(state["filters"], config["mode"])
