"""Small stat transformers.

ref: nodes.stats.RandomSignNode, LinearRectifier,
SignedHellingerMapper, NormalizeRows, ColumnSampler / Sampler.
All elementwise/row-local ops subclass ElementwiseTransformer so the
pipeline optimizer can fuse chains of them into one shard pass.
"""

from __future__ import annotations

import torch

from ...parallel.dataset import Dataset, TensorDataset
from ...workflow.optimizer import ElementwiseTransformer
from ...workflow.pipeline import Transformer


class RandomSignNode(ElementwiseTransformer):
    """Multiply each coordinate by a random fixed +-1 (ref: RandomSignNode)."""

    def __init__(self, dim: int, seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        self.dim = dim
        self.seed = seed
        self.signs = (torch.randint(0, 2, (dim,), generator=g,
                                    dtype=torch.int8).float() * 2 - 1)

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        self.signs = self.signs.to(device=X.device, dtype=X.dtype)
        return X * self.signs

    @classmethod
    def from_saved(cls, config, state):
        obj = cls.__new__(cls)
        obj.__dict__.update(config)
        obj.signs = state["signs"]
        return obj


class LinearRectifier(ElementwiseTransformer):
    """``clamp(x - alpha, min=max_val)`` = ``max(max_val, x - alpha)``
    (ref: LinearRectifier(maxVal, alpha) — same signature/semantics:
    shift by ``alpha``, floor at ``maxVal``)."""

    def __init__(self, max_val: float = 0.0, alpha: float = 0.0):
        self.max_val = max_val
        self.alpha = alpha

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        return torch.clamp(X - self.alpha, min=self.max_val)


class SignedHellingerMapper(ElementwiseTransformer):
    """``sign(x) * sqrt(|x|)`` — Fisher-Vector power normalization
    (ref: SignedHellingerMapper)."""

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        return torch.sign(X) * torch.sqrt(torch.abs(X))


class NormalizeRows(ElementwiseTransformer):
    """Row-wise L2 normalization (ref: NormalizeRows)."""

    def __init__(self, eps: float = 1e-12):
        self.eps = eps

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        return X / X.norm(dim=-1, keepdim=True).clamp_min(self.eps)


# ---- fused GPU fast path: SignedHellinger -> NormalizeRows is ONE HIP
# kernel (2 HBM passes instead of ~6; see hip/src/elementwise.hip)
def _hellinger_l2_gpu(parts, X: torch.Tensor) -> torch.Tensor:
    from ... import hip as hip_ops
    if X.dtype == torch.float32:
        return hip_ops.hellinger_l2(X, parts[1].eps)
    out = torch.sign(X) * torch.sqrt(torch.abs(X))
    return out / out.norm(dim=-1, keepdim=True).clamp_min(parts[1].eps)


def _register_fused():
    from ...workflow.optimizer import FUSED_GPU_PATTERNS
    FUSED_GPU_PATTERNS[(SignedHellingerMapper, NormalizeRows)] = _hellinger_l2_gpu


_register_fused()


class Sampler(Transformer):
    """Sample up to ``size`` items of the local shard (ref: Sampler)."""

    def __init__(self, size: int, seed: int = 42):
        self.size = size
        self.seed = seed

    def apply(self, x):
        return x

    def apply_dataset(self, ds: Dataset) -> Dataset:
        # `size` is the TOTAL sample budget across ranks (KeystoneML sampled
        # a global fraction); split it evenly.
        per_rank = max(1, self.size // max(1, ds.runtime.world_size))
        return ds.sample_local(per_rank, self.seed)


class ColumnSampler(Transformer):
    """From a dataset of [n_i, d] descriptor matrices, sample columns/rows to
    build a flat sample matrix (ref: ColumnSampler — used to draw ~1M SIFT
    descriptors for PCA/GMM fits)."""

    def __init__(self, num_samples: int, seed: int = 42):
        self.num_samples = num_samples
        self.seed = seed

    def apply(self, x: torch.Tensor) -> torch.Tensor:
        return x

    def apply_dataset(self, ds: Dataset) -> TensorDataset:
        per_rank = max(1, self.num_samples // max(1, ds.runtime.world_size))
        if isinstance(ds, TensorDataset) and ds.data.dim() >= 2:
            all_rows = ds.data.reshape(-1, ds.data.shape[-1])
        else:
            mats = [x for x in ds.items()]
            if not mats:
                raise ValueError("ColumnSampler: empty local shard")
            all_rows = torch.cat([m.reshape(-1, m.shape[-1]) for m in mats],
                                 dim=0)
        n = all_rows.shape[0]
        g = torch.Generator().manual_seed(self.seed + 7919 * ds.runtime.rank)
        if per_rank >= n:
            return TensorDataset(all_rows, ds.runtime)
        # with-replacement draw: a CPU randperm over millions of
        # descriptors costs ~100 ms; duplicates are harmless for the
        # PCA/GMM sample fits this feeds
        idx = torch.randint(0, n, (per_rank,), generator=g).to(all_rows.device)
        return TensorDataset(all_rows[idx], ds.runtime)
