"""Runtime configuration (config/flag system).

The reference used per-app scopt configs + Spark conf; here each app has a
dataclass config (models/*.py) and the runtime knobs live in ONE
``RuntimeConfig`` populated from ``KEYSTONE_*`` environment variables:

  KEYSTONE_HBM_BUDGET_GB   memory budget the optimizer plans against
  KEYSTONE_BLOCK_SIZE      default solver block size
  KEYSTONE_TRACE           JSONL trace output path (utils.tracing)
  KEYSTONE_ALLOW_EAGER     permit eager fallback where a HIP kernel exists
  KEYSTONE_COLL_TIMEOUT_S  RCCL/gloo collective timeout (abort cleanly)

Per-op debug / dispatch switches (read at the call site, not here):

  KEYSTONE_FV_GEMM=1       force the batched-GEMM FV path (skip fv_mfma)
  KEYSTONE_FV_KERNEL=1     use the v1 fused FV kernel (measured slower)
  KEYSTONE_GMM_EAGER=1     force the eager GMM E-step (skip STATS kernel)
  KEYSTONE_GMM_GRAPHS=1    hipGraph-capture the eager E-step (see gmm.py)
  KEYSTONE_SIFT_FUSED=1    fused SIFT pool kernel (measured slower; opt-in)
  KEYSTONE_FUSED_GEMM_COS=1  fully-fused MFMA gemm+cos (hybrid is faster)
  KEYSTONE_FV_F32STATS=1   exact-f32 stats pass in fv_mfma (default: f16
                           inputs / f32 accumulate on mfma_32x32x16_f16)
  KEYSTONE_FV_SPLIT=1      K-split 2-workgroups-per-image fv_mfma variant
                           (measured slower at K=256; kept as the A/B)
  KEYSTONE_NO_WBK=1        disable the fused per-class Woodbury kernel
                           (torch padded-bmm path instead)
  KEYSTONE_NO_PROFILE_CACHE=1  AutoCacheRule falls back to the
                           consumer-count heuristic (no sampled profiling)
  KEYSTONE_STATE_CACHE     fitted-estimator reuse directory (keys include
                           a fit-data fingerprint)
  KEYSTONE_GMM_DEBUG=1     per-phase timers in GMM/k-means fits
  KEYSTONE_SIFT_DEBUG=1    per-phase timers in SIFT extraction
  KEYSTONE_SOLVER_DEBUG=1  per-phase timers in the weighted block solver
  KEYSTONE_BENCH_STAGES=1  per-stage timings from bench.py (stderr JSON)
  KEYSTONE_NO_GRAPHS=1     disable hipGraph capture everywhere
"""

from __future__ import annotations

import os
from dataclasses import dataclass


@dataclass
class RuntimeConfig:
    hbm_budget_gb: float = 288.0
    block_size: int = 4096
    trace_path: str | None = None
    allow_eager: bool = False
    collective_timeout_s: int = 600

    @classmethod
    def from_env(cls) -> "RuntimeConfig":
        return cls(
            hbm_budget_gb=float(os.environ.get("KEYSTONE_HBM_BUDGET_GB", 288.0)),
            block_size=int(os.environ.get("KEYSTONE_BLOCK_SIZE", 4096)),
            trace_path=os.environ.get("KEYSTONE_TRACE"),
            allow_eager=bool(os.environ.get("KEYSTONE_ALLOW_EAGER")),
            collective_timeout_s=int(os.environ.get("KEYSTONE_COLL_TIMEOUT_S",
                                                    600)),
        )


_CONFIG: RuntimeConfig | None = None


def get_config() -> RuntimeConfig:
    global _CONFIG
    if _CONFIG is None:
        _CONFIG = RuntimeConfig.from_env()
    return _CONFIG
