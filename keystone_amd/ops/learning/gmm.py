"""Diagonal-covariance Gaussian mixture model (EM).

ref: nodes.learning.GaussianMixtureModelEstimator /
GaussianMixtureModel — the Fisher-Vector vocabulary model.  The reference
collected a descriptor sample to the driver and ran native EncEval
``computeGMM`` (C++); the rebuild runs *distributed* EM: the E-step
log-probabilities are a GEMM-shaped computation over the shard, and the
M-step reduces the sufficient statistics (Nk, sum gamma x, sum gamma x^2 —
K*(2D+1) floats) with one fused RCCL all-reduce per iteration.

MI355X path: the E-step + accumulation is the ``gmm_estep`` HIP kernel for
fp32 shards (keystone_amd/hip); torch fallback elsewhere.
"""

from __future__ import annotations

import math
import torch

from ...parallel.dataset import TensorDataset
from ...workflow.pipeline import BatchTransformer, Estimator
from .kmeans import KMeansPlusPlusEstimator

__all__ = ["GaussianMixtureModel", "GaussianMixtureModelEstimator"]


class GaussianMixtureModel(BatchTransformer):
    """Holds (means [K,D], variances [K,D], weights [K]); ``apply`` returns
    the responsibility (soft-assignment) matrix [n, K]."""

    def __init__(self, means: torch.Tensor, variances: torch.Tensor,
                 weights: torch.Tensor):
        self.means = means
        self.variances = variances
        self.weights = weights

    def log_prob(self, X: torch.Tensor) -> torch.Tensor:
        """[n, K] per-component log densities + log weights.
        Expanded-GEMM form: -(x^2) (1/2v) + x (mu/v) - mu^2/(2v) - norm."""
        mu = self.means.to(device=X.device, dtype=X.dtype)
        var = self.variances.to(device=X.device, dtype=X.dtype)
        w = self.weights.to(device=X.device, dtype=X.dtype)
        inv2v = 0.5 / var                                     # [K, D]
        a = (X * X) @ inv2v.T                                 # GEMM
        b = X @ (mu / var).T                                  # GEMM
        c = (mu * mu * inv2v).sum(1) + 0.5 * var.log().sum(1) \
            + 0.5 * mu.shape[1] * math.log(2 * math.pi)
        return -a + b - c + w.clamp_min(1e-30).log()

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        lp = self.log_prob(X)
        return torch.softmax(lp, dim=1)

    @classmethod
    def from_saved(cls, config, state):
        return cls(state["means"], state["variances"], state["weights"])


class GaussianMixtureModelEstimator(Estimator):
    def __init__(self, k: int, max_iterations: int = 50, seed: int = 0,
                 min_variance: float = 1e-6, tol: float = 1e-5,
                 init: str = "kmeans"):
        self.k = k
        self.max_iterations = max_iterations
        self.seed = seed
        self.min_variance = min_variance
        self.tol = tol
        self.init = init

    def _fit(self, data: TensorDataset) -> GaussianMixtureModel:
        import os
        import time
        dbg = os.environ.get("KEYSTONE_GMM_DEBUG")

        def tick(name, t0):
            if dbg:
                if data.data.is_cuda:
                    torch.cuda.synchronize()
                print(f"[gmm] {name}: {time.perf_counter() - t0:.4f}s",
                      flush=True)
            return time.perf_counter()

        t = time.perf_counter()
        X = data.data
        rt = data.runtime
        K, D = self.k, X.shape[1]
        dev, dt = X.device, X.dtype

        # ---- init: k-means centers + global variance
        if self.init == "kmeans":
            km = KMeansPlusPlusEstimator(K, max_iterations=5, seed=self.seed)._fit(data)
            mu = km.centers.to(dt)
            t = tick("kmeans_init", t)
        else:
            S = data.sample_local(max(2 * K // max(1, rt.world_size), K),
                                  self.seed).collect()
            g = torch.Generator().manual_seed(self.seed)
            mu = S[torch.randperm(S.shape[0], generator=g)[:K].to(S.device)].to(dt)
        stats = [torch.tensor([float(X.shape[0])], dtype=torch.float64, device=dev),
                 X.sum(0, dtype=torch.float64),
                 (X.to(torch.float64) ** 2).sum(0)]
        rt.all_reduce_many_(stats)
        n_tot = stats[0].item()
        gvar = (stats[2] / n_tot - (stats[1] / n_tot) ** 2).clamp_min(self.min_variance)
        var = gvar.unsqueeze(0).expand(K, D).to(dt).contiguous()
        w = torch.full((K,), 1.0 / K, dtype=dt, device=dev)

        t = tick("global_stats", t)
        model = GaussianMixtureModel(mu, var, w)
        # Fused E-step (fv_mfma.hip STATS mode): logits-MFMA -> softmax ->
        # moment-MFMA grid-strided over the shard, gamma [n, K] never hits
        # HBM — one launch replaces the 5-kernel eager chain (and the
        # [n, 2D] Xext hoist below is not needed at all).
        import os
        from ... import hip as hip_ops
        use_fused = (X.is_cuda and X.dtype == torch.float32
                     and hip_ops.gpu_ready()
                     and hip_ops.fv_mfma_supported(D, K)
                     and not os.environ.get("KEYSTONE_GMM_EAGER"))

        # Hoist the extended operand [X^2 | X] out of the EM loop: the
        # E-step logits become ONE addmm against the stacked coefficient
        # matrix [-1/2v ; mu/v] (2 HBM-resident copies of X buy ~half the
        # per-iteration kernel launches), and X2 is reused by Sxx.
        if not use_fused:
            X2 = X * X
            Xext = torch.cat([X2, X], dim=1)             # [n, 2D]

        # The per-iteration LOCAL compute (logit addmm -> softmax -> local
        # sufficient statistics) is a fixed launch-bound kernel chain —
        # capture it into ONE hipGraph and replay per iteration.  Only the
        # collective-free part is captured (the all-reduce and M-step stay
        # eager), so the behaviour is identical at every world size and no
        # RCCL call is ever captured.  KEYSTONE_NO_GRAPHS=1 disables.
        import os
        P_s = torch.empty(K, 2 * D, dtype=dt, device=dev)
        cc_s = torch.empty(K, dtype=dt, device=dev)

        def estep_local():
            lp = torch.addmm(-cc_s, Xext, P_s.T)
            lse = torch.logsumexp(lp, dim=1)
            gamma = torch.exp(lp - lse.unsqueeze(1))
            Nk_l = gamma.sum(0, dtype=torch.float64)
            # fp64 stats for fp64 inputs (oracle parity); fp32 GEMMs else
            if X.dtype == torch.float64:
                Sx_l = gamma.T @ X
                Sxx_l = gamma.T @ X2
            else:
                Sx_l = (gamma.T @ X).to(torch.float64)
                Sxx_l = (gamma.T @ X2).to(torch.float64)
            ll_l = lse.sum(dtype=torch.float64).reshape(1)
            return Nk_l, Sx_l, Sxx_l, ll_l

        def fill_coeffs(mdl):
            inv2v = 0.5 / mdl.variances
            cc_s.copy_((mdl.means * mdl.means * inv2v).sum(1)
                       + 0.5 * mdl.variances.log().sum(1)
                       + 0.5 * D * math.log(2 * math.pi)
                       - mdl.weights.clamp_min(1e-30).log())
            P_s[:, :D].copy_(-inv2v)
            P_s[:, D:].copy_(mdl.means / mdl.variances)

        # OPT-IN (KEYSTONE_GMM_GRAPHS=1): the captured E-step replays 3x
        # faster in isolation, but the graph's private memory pool
        # interacts pathologically with the multi-GB allocations of later
        # pipeline stages at large batches (FV encode measured 40-70x
        # slower afterwards, even with reset+empty_cache) — so the default
        # stays eager; revisit when torch's ROCm graph pools can be
        # reliably returned to the main allocator.
        graph = None
        if X.is_cuda and not use_fused and os.environ.get("KEYSTONE_GMM_GRAPHS"):
            try:
                fill_coeffs(model)
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    for _ in range(2):
                        estep_local()
                torch.cuda.current_stream().wait_stream(side)
                torch.cuda.synchronize()
                graph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(graph):
                    g_out = estep_local()
            except Exception:
                graph = None

        if use_fused:
            # two launches per EM iteration: the E-step partials kernel
            # (fv_mfma STATS mode) + the fused M-step/operand-build kernel
            # (gmm_mstep.hip) — the ~18 torch small-ops that rebuilt
            # (mu, var, w, P) each iteration were ~0.7 ms of launch
            # overhead, comparable to the E-step kernel itself.  This is
            # the answer to the shelved hipGraph capture:
            # same launch-count win, no graph memory-pool interaction.
            nb = min(512, max(1, (X.shape[0] + 31) // 32))
            width = K + 2 * K * D + 1
            st_out = torch.empty(nb, width, dtype=torch.float32, device=dev)
            st_P = torch.zeros(160, K, dtype=torch.float32, device=dev)
            mu_b = model.means.to(torch.float32).contiguous()
            var_b = model.variances.to(torch.float32).contiguous()
            w_b = model.weights.to(torch.float32).contiguous()
            ll_b = torch.zeros(1, dtype=torch.float32, device=dev)

            def fill_P(mdl):
                inv2v = 0.5 / mdl.variances
                st_P[:D].copy_(-inv2v.T)
                st_P[64:64 + D].copy_((mdl.means / mdl.variances).T)
                st_P[128].copy_(mdl.weights.clamp_min(1e-30).log()
                                - (mdl.means * mdl.means * inv2v).sum(1)
                                - 0.5 * mdl.variances.log().sum(1)
                                - 0.5 * D * math.log(2 * math.pi))

            fill_P(model)      # operand for iteration 0 (init model)
            prev_ll = -float("inf")
            for _it in range(self.max_iterations):
                hip_ops.gmm_stats_raw(X, st_P, st_out)
                # pre-reduce with torch's tree reduction: the kernel's
                # per-wave nb-loop measured 0.33 ms vs sum+mstep ~0.05
                pre = st_out.sum(0, dtype=torch.float64)
                if rt.is_distributed:
                    rt.all_reduce_(pre)
                hip_ops.gmm_mstep_(st_out, pre, float(n_tot),
                                   self.min_variance, mu_b, var_b, w_b,
                                   st_P, ll_b)
                if (_it & 3) == 3:
                    ll_now = ll_b.item() / n_tot
                    if abs(ll_now - prev_ll) < self.tol * max(1.0, abs(prev_ll)):
                        break
                    prev_ll = ll_now
            model = GaussianMixtureModel(mu_b.to(dt), var_b.to(dt),
                                         w_b.to(dt))
            tick(f"em_loop({_it + 1} iters)", t)
            return model

        prev_ll = -float("inf")
        for _it in range(self.max_iterations):
            if graph is not None:
                fill_coeffs(model)
                graph.replay()
                Nk = g_out[0].clone()
                Sx = g_out[1].clone()
                Sxx = g_out[2].clone()
                ll = g_out[3].clone()
            else:
                fill_coeffs(model)
                Nk, Sx, Sxx, ll = estep_local()
            rt.all_reduce_many_([Nk, Sx, Sxx, ll])
            Nk_c = Nk.clamp_min(1e-10).unsqueeze(1)
            mu64 = Sx / Nk_c
            var64 = (Sxx / Nk_c - mu64 ** 2).clamp_min(self.min_variance)
            model = GaussianMixtureModel(mu64.to(dt), var64.to(dt),
                                         (Nk / n_tot).to(dt))
            # convergence check every 4 iters: .item() is a device sync
            if (_it & 3) == 3:
                ll_now = ll.item() / n_tot
                if abs(ll_now - prev_ll) < self.tol * max(1.0, abs(prev_ll)):
                    break
                prev_ll = ll_now
        if graph is not None:
            # release the graph's private memory pool NOW: keeping it
            # reserved starves the multi-GB allocations of later pipeline
            # stages (measured: a leaked pool made the following FV encode
            # ~50x slower at bench scale from allocator thrash)
            del g_out
            graph.reset()
            del graph
        tick(f"em_loop({_it + 1} iters)", t)
        return model
