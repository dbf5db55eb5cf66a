"""Determinism / race-detection aids (aux subsystem).

The reference got determinism for free (immutable RDDs, functional
design).  Here the equivalents are: RCCL collectives are deterministic for
a fixed topology, every "driver" solve is replicated from all-reduced
inputs, and these helpers let tests/jobs ASSERT it:

  * ``model_fingerprint(t)`` — stable hash of a transformer's tensor state;
  * ``assert_replicated(x)`` — raises unless the value is bitwise identical
    on every rank (gather-based check, cheap for model-sized tensors);
  * ``assert_model_replicated(t)`` — the cross-rank model-hash equality
    check to run at epoch boundaries.
"""

from __future__ import annotations

import hashlib

import torch

from ..parallel.runtime import get_runtime

__all__ = ["model_fingerprint", "assert_replicated", "assert_model_replicated"]


def _tensor_digest(t: torch.Tensor) -> str:
    return hashlib.sha256(t.detach().cpu().contiguous().numpy().tobytes()).hexdigest()


def model_fingerprint(transformer) -> str:
    """Stable content hash over the transformer's persisted tensor state."""
    h = hashlib.sha256()
    state = transformer.get_state() if hasattr(transformer, "get_state") else {}
    for name in sorted(state):
        h.update(name.encode())
        h.update(_tensor_digest(state[name]).encode())
    return h.hexdigest()


def assert_replicated(x: torch.Tensor, name: str = "tensor") -> None:
    rt = get_runtime()
    if not rt.is_distributed:
        return
    digest = _tensor_digest(x)
    digests = [None] * rt.world_size
    import torch.distributed as dist
    dist.all_gather_object(digests, digest)
    if len(set(digests)) != 1:
        raise RuntimeError(
            f"determinism violation: {name} differs across ranks: {digests}")


def assert_model_replicated(transformer) -> None:
    rt = get_runtime()
    if not rt.is_distributed:
        return
    fp = model_fingerprint(transformer)
    fps = [None] * rt.world_size
    import torch.distributed as dist
    dist.all_gather_object(fps, fp)
    if len(set(fps)) != 1:
        raise RuntimeError(
            f"determinism violation: model fingerprints differ: {fps}")
