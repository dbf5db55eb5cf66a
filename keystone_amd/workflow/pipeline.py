"""Pipeline / Transformer / Estimator — the public composition API.

API-parity rebuild of KeystoneML's core types (ref: workflow package —
``Pipeline[A,B]``, ``Transformer[A,B]``, ``Estimator[A,B]``,
``LabelEstimator[A,B,L]``, ``PipelineDataset``, ``PipelineDatum``,
``Pipeline.gather`` — SURVEY.md §2.1):

  * ``Transformer`` — a pure function ``A -> B``; datum-level ``apply`` is
    mandatory, dataset lift is overridable (batched matrix ops override
    ``apply_batch`` and run one fused GPU launch per shard instead of a
    per-item map — the Spark ``mapPartitions`` idiom, made explicit because
    on a GPU every op is a kernel launch, not a free closure).
  * ``Estimator.fit(data) -> Transformer``; ``LabelEstimator.fit(data,
    labels)``.
  * ``Pipeline`` — a lazy DAG with one typed source and sink; composition
    with ``and_then`` (alias ``andThen``, operator ``>>``), fit-data splice
    ``and_then(est, data[, labels])``, branch/merge ``Pipeline.gather``.
  * ``PipelineDataset`` / ``PipelineDatum`` — lazy result handles; ``.get()``
    forces optimization + execution.

Every rank of the SPMD job builds the same pipeline object graph and
executes it identically on its own HBM-resident shard; collectives happen
inside estimators (Gram all-reduce etc.), never in the executor.
"""

from __future__ import annotations

from typing import Any, Callable, List, Optional, Sequence

import torch

from ..parallel.dataset import Dataset, ItemDataset, TensorDataset, from_items, from_rows
from .graph import (DatasetOperator, DatumOperator, DelegatingOperator,
                    EstimatorOperator, Graph, GraphId, NodeId, SourceId,
                    TransformerOperator)

__all__ = [
    "Pipeline", "Transformer", "BatchTransformer", "FunctionTransformer",
    "Estimator", "LabelEstimator", "PipelineDataset", "PipelineDatum",
    "Identity", "TupleDataset", "gather",
]


class TupleDataset(Dataset):
    """Zip of k datasets (the output of a gather node before combination)."""

    def __init__(self, parts: Sequence[Dataset]):
        super().__init__(parts[0].runtime if parts else None)
        self.parts = list(parts)

    def local_count(self) -> int:
        return self.parts[0].local_count()

    def items(self):
        return zip(*[p.items() for p in self.parts])

    def map(self, fn):
        return ItemDataset([fn(t) for t in self.items()], self.runtime)


# ==========================================================================
# Pipeline
# ==========================================================================
class PipelineBase:
    """Anything that can yield (graph, source, sink-dependency)."""

    def _graph_parts(self):
        raise NotImplementedError

    # -- composition -----------------------------------------------------
    def and_then(self, nxt, data=None, labels=None) -> "Pipeline":
        """``p.and_then(t)`` chains a transformer/pipeline;
        ``p.and_then(est, data[, labels])`` splices an estimator fitted on
        ``p(data)`` (ref: Pipeline.andThen overloads)."""
        if isinstance(nxt, (Estimator, LabelEstimator)) or data is not None:
            return self._and_then_estimator(nxt, data, labels)
        g1, s1, p1 = self._graph_parts()
        g2, s2, p2 = nxt._graph_parts()
        g = g1.copy()
        g.connect_graph(g2)
        g.replace_dependency(s2, p1)
        g.sources.remove(s2)
        return Pipeline(g, s1, p2)

    andThen = and_then

    def __rshift__(self, nxt):
        return self.and_then(nxt)

    def _and_then_estimator(self, est, data, labels=None) -> "Pipeline":
        if data is None:
            raise ValueError("and_then(estimator, data[, labels]) requires fit data")
        g1, s1, p1 = self._graph_parts()
        g = g1.copy()
        data_node = g.add_node(DatasetOperator(_as_dataset(data)), [])
        # Duplicate this pipeline's chain with the source replaced by the fit
        # data: the fit path and the transform path share operator objects, so
        # the optimizer's node-merge rule can CSE them when profitable.
        mapping = _duplicate_chain(g, p1, {s1: data_node})
        fit_input = mapping.get(p1, p1)
        deps: List[GraphId] = [fit_input]
        if labels is not None:
            deps.append(g.add_node(DatasetOperator(_as_dataset(labels)), []))
        est_node = g.add_node(EstimatorOperator(est), deps)
        apply_node = g.add_node(DelegatingOperator(), [est_node, p1])
        return Pipeline(g, s1, apply_node)

    # -- application -----------------------------------------------------
    def apply(self, inp):
        g, s, p = self._graph_parts()
        g = g.copy()
        if isinstance(inp, Dataset):
            g.replace_source_with_node(s, DatasetOperator(inp))
            sink = g.add_sink(p)
            return PipelineDataset(g, sink)
        g.replace_source_with_node(s, DatumOperator(inp))
        sink = g.add_sink(p)
        return PipelineDatum(g, sink)

    def __call__(self, inp):
        return self.apply(inp)

    # -- training --------------------------------------------------------
    def fit(self) -> "Pipeline":
        """Force every estimator fit in the DAG and return a
        transformer-only pipeline (what ``Pipeline.save`` persists)."""
        from .executor import fit_pipeline
        return fit_pipeline(self)

    # -- persistence (ref: Pipeline save/load) -----------
    def save(self, path: str) -> None:
        from .serialization import save_pipeline
        save_pipeline(self, path)

    @staticmethod
    def load(path: str) -> "Pipeline":
        from .serialization import load_pipeline
        return load_pipeline(path)


class Pipeline(PipelineBase):
    """A concrete lazy pipeline: graph + input source + output dependency."""

    def __init__(self, graph: Graph, source: SourceId, out: GraphId):
        self.graph = graph
        self.source = source
        self.out = out

    def _graph_parts(self):
        return self.graph, self.source, self.out

    # introspection helpers (used by tests / optimizer)
    def transformers(self) -> List["Transformer"]:
        return [op.transformer for op in self.graph.operators.values()
                if isinstance(op, TransformerOperator)]

    @staticmethod
    def gather(branches: Sequence[PipelineBase]) -> "Pipeline":
        return gather(branches)


def gather(branches: Sequence[PipelineBase]) -> Pipeline:
    """``gather([p1..pk]) : A -> (B1..Bk)`` — apply every branch to the same
    input and tuple the results (ref: Pipeline.gather).  Follow with
    ``VectorCombiner`` to concatenate feature branches."""
    if not branches:
        raise ValueError("gather() needs at least one branch")
    g = Graph()
    s = g.add_source()
    outs = []
    for b in branches:
        gb, sb, pb = b._graph_parts()
        g.connect_graph(gb)
        g.replace_dependency(sb, s)
        g.sources.remove(sb)
        outs.append(pb)
    node = g.add_node(TransformerOperator(_GatherTransformer(len(branches))), outs)
    return Pipeline(g, s, node)


def _duplicate_chain(g: Graph, target: GraphId, subst: dict) -> dict:
    """Copy the sub-DAG feeding ``target`` into ``g``, replacing ids in
    ``subst``; returns old-id -> new-id mapping (data nodes are shared)."""
    mapping = dict(subst)

    def visit(gid: GraphId) -> GraphId:
        if gid in mapping:
            return mapping[gid]
        if not isinstance(gid, NodeId):
            mapping[gid] = gid
            return gid
        op = g.operators[gid]
        if isinstance(op, (DatasetOperator, DatumOperator)):
            mapping[gid] = gid  # bound data is shared, not copied
            return gid
        new_deps = [visit(d) for d in g.dependencies[gid]]
        if new_deps == g.dependencies[gid]:
            mapping[gid] = gid  # nothing upstream changed: reuse
            return gid
        nid = g.add_node(op, new_deps)
        mapping[gid] = nid
        return nid

    visit(target)
    return mapping


def _as_dataset(x) -> Dataset:
    if isinstance(x, Dataset):
        return x
    if isinstance(x, torch.Tensor):
        return from_rows(x)
    if isinstance(x, (list, tuple)):
        return from_items(x)
    raise TypeError(f"cannot treat {type(x)} as a Dataset")


# ==========================================================================
# Lazy result handles
# ==========================================================================
class _LazyResult:
    def __init__(self, graph: Graph, sink):
        self.graph = graph
        self.sink = sink
        self._value = None
        self._done = False

    def get(self):
        if not self._done:
            from .executor import execute_sink
            self._value = execute_sink(self.graph, self.sink)
            self._done = True
        return self._value


class PipelineDataset(_LazyResult):
    """Lazy handle on a distributed result (ref: PipelineDataset)."""

    def collect(self):
        return self.get().collect()


class PipelineDatum(_LazyResult):
    """Lazy handle on a single-datum result (ref: PipelineDatum)."""


# ==========================================================================
# Transformer / Estimator
# ==========================================================================
class Transformer(PipelineBase):
    """A pure function ``A -> B``, liftable over datasets; itself a Pipeline
    (ref: Transformer extends Pipeline)."""

    def apply(self, x):
        """Datum-level application — mandatory unless apply_batch is given."""
        ab = getattr(self, "apply_batch", None)
        if ab is not None and isinstance(x, torch.Tensor):
            return ab(x.unsqueeze(0)).squeeze(0)
        raise NotImplementedError(f"{type(self).__name__}.apply")

    def apply_dataset(self, ds: Dataset) -> Dataset:
        """Dataset lift; default is per-item map, batched ops override
        (directly or via ``apply_batch``)."""
        ab = getattr(self, "apply_batch", None)
        if ab is not None and isinstance(ds, TensorDataset):
            return ds.map_batch(ab)
        return ds.map(self.apply)

    def _graph_parts(self):
        g = Graph()
        s = g.add_source()
        n = g.add_node(TransformerOperator(self), [s])
        return g, s, n

    def __call__(self, inp):
        # Convenience: direct eager call on a raw tensor / datum, lazy handle
        # on a Dataset (keeps the Pipeline contract for datasets).
        if isinstance(inp, Dataset):
            return self.apply_dataset(inp)
        if isinstance(inp, (PipelineDataset, PipelineDatum)):
            return self.apply(inp.get())
        return self.apply(inp)

    # -- persistence hooks (overridden by stateful/fitted transformers) --
    def get_state(self) -> dict:
        """Tensors to persist (safetensors payload). Override if stateful."""
        return {t: v for t, v in vars(self).items() if isinstance(v, torch.Tensor)}

    def get_config(self) -> dict:
        """JSON-serializable hyperparameters. Defaults to plain attrs."""
        out = {}
        for k, v in vars(self).items():
            if isinstance(v, (int, float, str, bool, type(None))):
                out[k] = v
            elif isinstance(v, (list, tuple)) and all(
                    isinstance(e, (int, float, str, bool)) for e in v):
                out[k] = list(v)
        return out

    @classmethod
    def from_saved(cls, config: dict, state: dict) -> "Transformer":
        obj = cls.__new__(cls)
        obj.__dict__.update(config)
        obj.__dict__.update(state)
        return obj


class BatchTransformer(Transformer):
    """Base for ops whose natural unit is the whole shard tensor."""

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def apply(self, x):
        return self.apply_batch(x.unsqueeze(0)).squeeze(0)

    def apply_dataset(self, ds: Dataset) -> Dataset:
        if isinstance(ds, TensorDataset):
            return ds.map_batch(self.apply_batch)
        return ds.map(self.apply)


class FunctionTransformer(Transformer):
    """Wrap a plain function as a Transformer (ref: Transformer.apply sugar)."""

    def __init__(self, fn: Callable, name: Optional[str] = None):
        self.fn = fn
        self.name = name or getattr(fn, "__name__", "fn")

    def apply(self, x):
        return self.fn(x)


class Identity(Transformer):
    """ref: nodes.util.Identity"""

    def apply(self, x):
        return x

    def apply_dataset(self, ds):
        return ds


class _GatherTransformer(Transformer):
    """Internal: tuple the outputs of gather branches."""

    def __init__(self, k: int):
        self.k = k

    def apply(self, *xs):
        return tuple(xs)

    def apply_dataset(self, *parts: Dataset) -> TupleDataset:
        return TupleDataset(parts)


class Estimator:
    """``fit(data) -> Transformer`` (ref: Estimator)."""

    def fit(self, data) -> Transformer:
        return self._fit(_as_dataset(data))

    def _fit(self, data: Dataset) -> Transformer:
        raise NotImplementedError

    def with_data(self, data) -> Pipeline:
        """An unfitted pipeline stage: identity chained with this estimator
        (ref: Estimator.withData)."""
        return Identity().and_then(self, data)


class LabelEstimator:
    """``fit(data, labels) -> Transformer`` (ref: LabelEstimator)."""

    def fit(self, data, labels) -> Transformer:
        return self._fit(_as_dataset(data), _as_dataset(labels))

    def _fit(self, data: Dataset, labels: Dataset) -> Transformer:
        raise NotImplementedError

    def with_data(self, data, labels) -> Pipeline:
        return Identity().and_then(self, data, labels)
