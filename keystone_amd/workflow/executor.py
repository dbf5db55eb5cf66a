"""Graph executor: lazy topological execution with memoization.

ref: workflow GraphExecutor / Expression.  Executes a
graph to a sink, memoizing intermediate expressions; the whole-pipeline
optimizer rewrites the graph first.  In the MI355X runtime every rank runs
this executor identically over its HBM shard (SPMD) — there is no
driver/executor split and no task scheduling: operators are either fused
tensor ops / HIP kernel launches (shard-level) or per-item maps.

The executor records a per-node profile (wall seconds, output bytes) that
feeds the greedy AutoCacheRule (ICDE'17-style whole-pipeline optimization)
and the optional JSONL trace (utils.tracing).
"""

from __future__ import annotations

import time
from typing import Any, Dict

import torch

from ..parallel.dataset import Dataset, ItemDataset, TensorDataset
from ..utils.tracing import trace_event
from .graph import (DatasetOperator, DatumOperator, DelegatingOperator,
                    EstimatorOperator, Graph, GraphId, NodeId, SinkId,
                    TransformerOperator)
from .pipeline import Pipeline, PipelineBase, Transformer

__all__ = ["GraphExecutor", "execute_sink", "fit_pipeline"]


def _nbytes(value: Any) -> int:
    if isinstance(value, torch.Tensor):
        return value.numel() * value.element_size()
    if isinstance(value, TensorDataset):
        return _nbytes(value.data)
    if isinstance(value, ItemDataset):
        return sum(_nbytes(x) for x in list(value.items())[:16]) * max(1, value.local_count() // 16) \
            if value.local_count() else 0
    return 0


class GraphExecutor:
    """Executes one (already-bound) graph; memoizes node results."""

    def __init__(self, graph: Graph, optimize: bool = True):
        if optimize:
            from .optimizer import default_optimizer
            graph = default_optimizer().execute(graph)
        self.graph = graph
        self.results: Dict[GraphId, Any] = {}
        self.profile: Dict[NodeId, dict] = {}

    def execute(self, gid: GraphId) -> Any:
        if gid in self.results:
            return self.results[gid]
        if gid in self.graph.sinks:
            value = self.execute(self.graph.sinks[gid])
            self.results[gid] = value
            return value
        if not isinstance(gid, NodeId):
            raise RuntimeError(f"unbound source {gid!r}: apply the pipeline to data first")

        op = self.graph.operators[gid]
        deps = self.graph.dependencies[gid]
        inputs = [self.execute(d) for d in deps]

        t0 = time.perf_counter()
        value = self._run(op, inputs)
        dt = time.perf_counter() - t0
        self.profile[gid] = {"op": op.label, "seconds": dt, "bytes": _nbytes(value)}
        trace_event("op", name=op.label, seconds=dt, bytes=self.profile[gid]["bytes"])

        self.results[gid] = value
        return value

    # ------------------------------------------------------------------
    def _run(self, op, inputs):
        if isinstance(op, DatasetOperator):
            return op.dataset
        if isinstance(op, DatumOperator):
            return op.datum
        if isinstance(op, TransformerOperator):
            return self._apply_transformer(op.transformer, inputs)
        if isinstance(op, DelegatingOperator):
            fitted = inputs[0]
            if not isinstance(fitted, Transformer):
                raise RuntimeError("DelegatingOperator dep[0] did not produce a Transformer")
            return self._apply_transformer(fitted, inputs[1:])
        if isinstance(op, EstimatorOperator):
            est = op.estimator
            # Node-level operator selection (ref: Optimizable{Estimator,...}
            # optimize(sample, numPerPartition)): cost-model physical choice.
            from .optimizer import Optimizable
            if isinstance(est, Optimizable):
                est = est.optimize(*inputs)
            # fitted-state reuse (ref: SavedStateLoadRule): key by estimator
            # class + hyperparameters when a state cache is enabled
            from .state_cache import (data_fingerprint, save_fitted,
                                      state_cache_dir, try_load)
            if state_cache_dir() is not None:
                # key by (class, hyperparams, fit-data fingerprint): identical
                # estimators fit on different datasets must not alias
                tag = data_fingerprint(*inputs[:2])
                cached = try_load(est, tag)
                if cached is not None:
                    return cached
            fitted = est.fit(*inputs[:2])
            if state_cache_dir() is not None:
                save_fitted(est, fitted, tag)
            return fitted
        raise RuntimeError(f"unknown operator {op!r}")

    @staticmethod
    def _apply_transformer(t: Transformer, inputs):
        if any(isinstance(x, Dataset) for x in inputs):
            return t.apply_dataset(*inputs)
        return t.apply(*inputs)


def execute_sink(graph: Graph, sink: SinkId):
    return GraphExecutor(graph).execute(sink)


def fit_pipeline(pipeline: PipelineBase) -> Pipeline:
    """Force every estimator fit; return a transformer-only pipeline.

    Estimator nodes never depend on the pipeline's unbound source (their
    data was bound by ``and_then(est, data)``), so they can be executed
    now; each (EstimatorOperator -> DelegatingOperator) pair collapses to a
    fitted TransformerOperator.  ref: KeystoneML's lazy fit + FittedPipeline.
    """
    g, source, out = pipeline._graph_parts()
    g = g.copy()
    ex = GraphExecutor(g)
    g = ex.graph  # post-optimization graph

    new_g = g.copy()
    for nid in list(g.operators):
        op = g.operators[nid]
        if isinstance(op, DelegatingOperator):
            est_dep = g.dependencies[nid][0]
            fitted = ex.execute(est_dep)
            new_g.operators[nid] = TransformerOperator(fitted)
            new_g.dependencies[nid] = g.dependencies[nid][1:]

    # prune nodes no longer reachable from the output
    keep = set(new_g.ancestors(out)) | {out}
    new_g.operators = {n: o for n, o in new_g.operators.items() if n in keep}
    new_g.dependencies = {n: d for n, d in new_g.dependencies.items() if n in keep}
    new_g.sinks = {}
    new_g.sources = [s for s in new_g.sources if s in keep or s == source]
    if source not in new_g.sources:
        new_g.sources.append(source)
    return Pipeline(new_g, source, out)
