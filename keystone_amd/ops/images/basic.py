"""Basic image transformers.

ref: nodes.images.GrayScaler, PixelScaler,
ImageVectorizer, Windower / FastWindower, Pooler, SymmetricRectifier.

Image convention: a datum is ``[H, W]`` (gray) or ``[C, H, W]``; the
batched shard form adds a leading B dim.  All ops are written batched
(one kernel chain per shard, not per image).
"""

from __future__ import annotations

from typing import Callable, Optional

import torch
import torch.nn.functional as F

from ...workflow.optimizer import ElementwiseTransformer
from ...workflow.pipeline import BatchTransformer, Transformer


class GrayScaler(BatchTransformer):
    """RGB -> luminance gray (ref: GrayScaler). [B?, 3, H, W] -> [B?, H, W];
    already-gray inputs pass through."""

    def __init__(self):
        self._w = torch.tensor([0.299, 0.587, 0.114])

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        if X.dim() == 3:  # [B, H, W] already gray
            return X
        # cache weights on the compute device (a per-call H2D upload would
        # also break hipGraph capture of the serving path)
        self._w = self._w.to(device=X.device, dtype=X.dtype)
        if X.shape[1] == 1:
            return X[:, 0]
        # explicit weighted sum: the einsum form lowers to a degenerate
        # [B*H*W, 3] @ [3, 1] GEMM that hipBLASLt rejects
        # (HIPBLAS_STATUS_NOT_SUPPORTED) and a slow fallback runs instead
        # — 23.2 ms vs ~2 ms for these elementwise passes (8192x3x256x256)
        w = self._w
        return X[:, 0] * w[0] + X[:, 1] * w[1] + X[:, 2] * w[2]

    def apply(self, x):
        if x.dim() == 2:
            return x
        return self.apply_batch(x.unsqueeze(0)).squeeze(0)


class PixelScaler(ElementwiseTransformer):
    """uint8 pixel range -> [0,1] float (ref: PixelScaler)."""

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        return X.float() / 255.0


class ImageVectorizer(BatchTransformer):
    """image -> flat vector (ref: ImageVectorizer)."""

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        return X.reshape(X.shape[0], -1)

    def apply(self, x):
        return x.reshape(-1)


class Windower(Transformer):
    """Slide a window over the image producing the patch stream
    (ref: Windower / FastWindower).  [C, H, W] -> [n_patches, C*size*size]."""

    def __init__(self, stride: int, size: int):
        self.stride = stride
        self.size = size

    def apply(self, x: torch.Tensor) -> torch.Tensor:
        if x.dim() == 2:
            x = x.unsqueeze(0)
        patches = F.unfold(x.unsqueeze(0), self.size, stride=self.stride)
        return patches.squeeze(0).T.contiguous()  # [n, C*size*size]

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        """[B, C?, H, W] -> [B, n_patches, C*size*size]"""
        if X.dim() == 3:
            X = X.unsqueeze(1)
        return F.unfold(X, self.size, stride=self.stride).transpose(1, 2).contiguous()


class FastWindower(Transformer):
    """Same patch stream as Windower via a zero-copy as_strided view
    (ref: FastWindower): no im2col kernel — the gather happens once at
    the consumer's reshape instead of materializing inside the op."""

    def __init__(self, stride: int, size: int):
        self.stride = stride
        self.size = size

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        """[B, C?, H, W] -> [B, n_patches, C*size*size]"""
        if X.dim() == 3:
            X = X.unsqueeze(1)
        B, C, H, W = X.shape
        k, st = self.size, self.stride
        ny = (H - k) // st + 1
        nx = (W - k) // st + 1
        s0, s1, s2, s3 = X.stride()
        v = X.as_strided((B, ny, nx, C, k, k),
                         (s0, st * s2, st * s3, s1, s2, s3))
        return v.reshape(B, ny * nx, C * k * k)

    def apply(self, x: torch.Tensor) -> torch.Tensor:
        if x.dim() == 2:
            x = x.unsqueeze(0)
        return self.apply_batch(x.unsqueeze(0)).squeeze(0)


class Pooler(BatchTransformer):
    """Spatial pooling with stride/size and a pixel function
    (ref: Pooler(stride, poolSize, pixelFunc, poolFunc) — the CIFAR
    random-patch pipeline pools rectified conv maps)."""

    def __init__(self, stride: int, size: int, mode: str = "sum",
                 pixel_fn: Optional[Callable] = None):
        self.stride = stride
        self.size = size
        self.mode = mode
        self.pixel_fn = pixel_fn

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        # X: [B, F, H, W]
        if self.pixel_fn is not None:
            X = self.pixel_fn(X)
        if self.mode == "max":
            return F.max_pool2d(X, self.size, self.stride)
        out = F.avg_pool2d(X, self.size, self.stride)
        return out * (self.size * self.size)

    def apply(self, x):
        return self.apply_batch(x.unsqueeze(0)).squeeze(0)

    def get_config(self):
        return {"stride": self.stride, "size": self.size, "mode": self.mode}

    @classmethod
    def from_saved(cls, config, state):
        return cls(config["stride"], config["size"], config["mode"])


class SymmetricRectifier(BatchTransformer):
    """x -> [max(0, x-alpha); max(0, -x-alpha)] feature doubling along the
    channel/feature dim (ref: SymmetricRectifier).  GPU fp32 path: one
    fused HIP kernel."""

    def __init__(self, alpha: float = 0.0):
        self.alpha = alpha

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        if X.is_cuda and X.dtype == torch.float32 and X.dim() == 2:
            from ... import hip as hip_ops
            if hip_ops.gpu_ready():
                return hip_ops.sym_rectify(X, self.alpha)
        return torch.cat([torch.clamp(X - self.alpha, min=0),
                          torch.clamp(-X - self.alpha, min=0)], dim=1)

    def apply(self, x):
        return torch.cat([torch.clamp(x - self.alpha, min=0),
                          torch.clamp(-x - self.alpha, min=0)], dim=0)
