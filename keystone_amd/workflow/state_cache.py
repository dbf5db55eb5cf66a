"""Fitted-state reuse across runs.

ref (SURVEY.md §2.1 [L]): workflow.SavedStateLoadRule /
ExtractSaveablePrefixes — the optimizer splices previously persisted
intermediates into a new run.  Rebuild: when a state cache is enabled,
the executor keys every estimator fit by (class, hyperparameters, user
tag); a hit loads the previously fitted transformer from disk instead of
refitting (the transformer payload uses the standard manifest+safetensors
format of workflow.serialization).
"""

from __future__ import annotations

import hashlib
import json
import os
from typing import Optional

_STATE_DIR: Optional[str] = None


def enable_state_cache(path: str) -> None:
    global _STATE_DIR
    os.makedirs(path, exist_ok=True)
    _STATE_DIR = path


def disable_state_cache() -> None:
    global _STATE_DIR
    _STATE_DIR = None


def state_cache_dir() -> Optional[str]:
    return _STATE_DIR or os.environ.get("KEYSTONE_STATE_CACHE")


def _estimator_key(est, tag: str = "") -> str:
    cfg = {}
    for k, v in vars(est).items():
        if isinstance(v, (int, float, str, bool, type(None))):
            cfg[k] = v
    blob = json.dumps({"cls": type(est).__module__ + "." + type(est).__qualname__,
                       "cfg": cfg, "tag": tag}, sort_keys=True)
    return hashlib.sha256(blob.encode()).hexdigest()[:24]


def data_fingerprint(*inputs) -> str:
    """Cheap content fingerprint of the fit data, folded into the cache key
    so two estimators with identical hyperparameters fit on different
    datasets never alias (shape + dtype + sum + a strided element sample
    per tensor input; datum inputs fall back to repr)."""
    import torch
    parts = []
    for x in inputs:
        t = getattr(x, "data", x)  # TensorDataset -> shard tensor
        if isinstance(t, torch.Tensor):
            flat = t.detach().reshape(-1)
            k = min(64, flat.numel())
            idx = torch.linspace(0, max(0, flat.numel() - 1), k,
                                 device=flat.device).long() if k else None
            sample = flat[idx].to(torch.float64).cpu().numpy().tobytes() \
                if k else b""
            parts.append((tuple(t.shape), str(t.dtype),
                          float(flat.to(torch.float64).sum().cpu()),
                          hashlib.sha256(sample).hexdigest()[:16]))
        else:
            parts.append(repr(t)[:256])
    return hashlib.sha256(repr(parts).encode()).hexdigest()[:16]


def try_load(est, tag: str = ""):
    d = state_cache_dir()
    if d is None:
        return None
    path = os.path.join(d, _estimator_key(est, tag))
    if not os.path.exists(os.path.join(path, "manifest.json")):
        return None
    from .serialization import load_pipeline
    from .graph import TransformerOperator
    pipe = load_pipeline(path)
    # single-node pipeline -> the fitted transformer itself
    ops = [op for op in pipe.graph.operators.values()
           if isinstance(op, TransformerOperator)]
    return ops[0].transformer if len(ops) == 1 else None


def save_fitted(est, fitted, tag: str = "") -> None:
    d = state_cache_dir()
    if d is None:
        return
    from .serialization import save_pipeline
    path = os.path.join(d, _estimator_key(est, tag))
    save_pipeline(_single(fitted), path)


def _single(t):
    from .graph import Graph, TransformerOperator
    from .pipeline import Pipeline
    g = Graph()
    s = g.add_source()
    n = g.add_node(TransformerOperator(t), [s])
    return Pipeline(g, s, n)
