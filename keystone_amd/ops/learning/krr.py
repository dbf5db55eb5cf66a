"""Blocked kernel ridge regression.

ref: nodes.learning.KernelRidgeRegression,
KernelBlockLinearMapper, GaussianKernelGenerator, KernelMatrix /
BlockKernelMatrix (KeystoneML v0.4+).

Solves ``alpha = (K + lam * n I)^{-1} Y`` by block coordinate descent over
blocks of *training points* (dual BCD — the primal feature blocking of
BlockLeastSquares transposed to the kernel side):

  per epoch, per block b of rows:
    * the block's points X_b are all-gathered (the heavy xGMI traffic);
    * every rank computes its slice of K(X_b, X_local) @ alpha_local and the
      block residual is all-reduced;
    * the owner solves the small (B x B) system (K_bb + lam n I) and the
      updated alpha_b is broadcast (here: replicated solve on all ranks —
      K_bb and the residual are already replicated after the reduce).

The fitted ``KernelBlockLinearMapper`` keeps the training shard + its alpha
slice on each rank; ``apply`` computes K(test, X_local) @ alpha_local and
all-reduces.
"""

from __future__ import annotations


import torch

from ...parallel.dataset import Dataset, TensorDataset
from ...workflow.pipeline import LabelEstimator, Transformer

__all__ = ["GaussianKernelGenerator", "KernelRidgeRegression",
           "KernelBlockLinearMapper"]


class GaussianKernelGenerator:
    """K(x, z) = exp(-gamma ||x - z||^2) (ref: GaussianKernelGenerator)."""

    def __init__(self, gamma: float):
        self.gamma = gamma

    def __call__(self, A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
        d2 = (A * A).sum(1, keepdim=True) - 2.0 * (A @ B.T) + (B * B).sum(1)
        return torch.exp(-self.gamma * d2.clamp_min(0))


class KernelBlockLinearMapper(Transformer):
    """Distributed kernel predictor: each rank holds (X_local, alpha_local)."""

    def __init__(self, kernel: GaussianKernelGenerator, X_local: torch.Tensor,
                 alpha_local: torch.Tensor, runtime=None):
        from ...parallel.runtime import get_runtime
        self.kernel = kernel
        self.X_local = X_local
        self.alpha_local = alpha_local
        self.runtime = runtime or get_runtime()

    def apply(self, x: torch.Tensor) -> torch.Tensor:
        out = self.apply_batch(x.unsqueeze(0) if x.dim() == 1 else x)
        return out.squeeze(0) if x.dim() == 1 else out

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        # NOTE: test rows are assumed REPLICATED across ranks here (each
        # rank computes its kernel slice and the all-reduce completes the
        # sum over training points).
        out = self.kernel(X, self.X_local.to(X.device)) @ \
            self.alpha_local.to(device=X.device, dtype=X.dtype)
        self.runtime.all_reduce_(out)
        return out

    def apply_dataset(self, ds: Dataset):
        if isinstance(ds, TensorDataset):
            # sharded test rows: every rank needs kernel against ALL training
            # points; all-gather each rank's test shard in turn.
            rt = ds.runtime
            if not rt.is_distributed:
                return ds.map_batch(self.apply_batch)
            X_all = rt.all_gather_rows(ds.data)
            out_all = self.kernel(X_all, self.X_local) @ self.alpha_local
            rt.all_reduce_(out_all)
            lo, hi = rt.shard_bounds(X_all.shape[0])
            return TensorDataset(out_all[lo:hi], rt)
        return ds.map(self.apply)

    def get_state(self):
        return {"X_local": self.X_local, "alpha_local": self.alpha_local,
                "gamma": torch.tensor(self.kernel.gamma)}

    @classmethod
    def from_saved(cls, config, state):
        return cls(GaussianKernelGenerator(float(state["gamma"].item())),
                   state["X_local"], state["alpha_local"])


class KernelRidgeRegression(LabelEstimator):
    def __init__(self, kernel: GaussianKernelGenerator, lam: float = 1e-3,
                 block_size: int = 2048, num_epochs: int = 2):
        self.kernel = kernel
        self.lam = lam
        self.block_size = block_size
        self.num_epochs = num_epochs

    def _fit(self, data: TensorDataset, labels: TensorDataset) -> KernelBlockLinearMapper:
        X, Y = data.data, labels.data
        if Y.dim() == 1:
            Y = Y.unsqueeze(1)
        Y = Y.to(X.dtype)
        rt = data.runtime
        n_local = X.shape[0]
        counts = rt.all_gather(torch.tensor([n_local], dtype=torch.int64))
        counts = [int(c.item()) for c in counts]
        offsets = [sum(counts[:r]) for r in range(len(counts))]
        n_tot = sum(counts)
        alpha = torch.zeros(n_local, Y.shape[1], dtype=X.dtype, device=X.device)

        # global block index ranges over the concatenated row ordering
        blocks = []
        lo = 0
        while lo < n_tot:
            blocks.append((lo, min(lo + self.block_size, n_tot)))
            lo += self.block_size

        my_lo = offsets[rt.rank]
        my_hi = my_lo + n_local
        lam_n = self.lam * n_tot

        for _epoch in range(self.num_epochs):
            for (lo, hi) in blocks:
                # owner ranks contribute their rows of the block
                sl = slice(max(lo, my_lo) - my_lo, max(min(hi, my_hi), my_lo) - my_lo)
                Xb_part = X[sl]
                Yb_part = Y[sl]
                Xb = rt.all_gather_rows(TensorDataset(Xb_part, rt).data) \
                    if rt.is_distributed else Xb_part
                Yb = rt.all_gather_rows(TensorDataset(Yb_part, rt).data) \
                    if rt.is_distributed else Yb_part
                # block prediction: K(X_b, X_all) alpha  (distributed sum)
                Kb_local = self.kernel(Xb, X)                      # [B, n_local]
                pred = Kb_local @ alpha
                rt.all_reduce_(pred)
                # current alpha_b (gathered)
                ab = rt.all_gather_rows(TensorDataset(alpha[sl], rt).data) \
                    if rt.is_distributed else alpha[sl]
                K_bb = self.kernel(Xb, Xb)
                # exact block coordinate update of (K + lam n I) alpha = Y:
                # (K_bb + lam n I) a_b_new = Y_b - K_{b,:} alpha + K_bb a_b
                # (pred contains K_bb a_b_old, so add it back; the lam_n
                # term belongs ONLY on the left side)
                resid = Yb - pred + K_bb @ ab
                # solve (K_bb + lam n I) a_b = resid  (replicated, fp64)
                A = K_bb.to(torch.float64) + lam_n * torch.eye(
                    Xb.shape[0], dtype=torch.float64, device=Xb.device)
                ab_new = torch.linalg.solve(A, resid.to(torch.float64)).to(X.dtype)
                # scatter back the rows this rank owns
                blk_off = max(lo, my_lo) - lo
                alpha[sl] = ab_new[blk_off: blk_off + (sl.stop - sl.start)]
        return KernelBlockLinearMapper(self.kernel, X, alpha, rt)
