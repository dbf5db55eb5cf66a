"""Graph IR for pipelines.

Mirrors KeystoneML's workflow graph (ref: workflow/graph — ``Graph``,
``Operator``, ``GraphId``/``NodeId``/``SourceId``/``SinkId``; operators
``TransformerOperator`` / ``EstimatorOperator`` / ``DatasetOperator`` /
``DatumOperator`` / ``DelegatingOperator``), rebuilt in
Python.  The graph is a DAG:

  * **sources** — typed holes filled at apply time (the pipeline input);
  * **nodes**   — each carries an ``Operator`` and an ordered dependency
                  list of graph ids (sources or other nodes);
  * **sinks**   — named outputs pointing at a source or node.

Unlike the reference this IR never sees Spark: executing an operator hands
it *Expressions* over HBM-resident shards (see executor.py).
"""

from __future__ import annotations

import itertools
from dataclasses import dataclass, field
from typing import Dict, List, Tuple

__all__ = [
    "GraphId", "NodeId", "SourceId", "SinkId", "Graph", "Operator",
    "TransformerOperator", "EstimatorOperator", "DelegatingOperator",
    "DatasetOperator", "DatumOperator",
]

_counter = itertools.count(1)


@dataclass(frozen=True)
class GraphId:
    idx: int

    def __repr__(self):
        return f"{type(self).__name__}({self.idx})"


class NodeId(GraphId):
    pass


class SourceId(GraphId):
    pass


class SinkId(GraphId):
    pass


def fresh_node_id() -> NodeId:
    return NodeId(next(_counter))


def fresh_source_id() -> SourceId:
    return SourceId(next(_counter))


def fresh_sink_id() -> SinkId:
    return SinkId(next(_counter))


# --------------------------------------------------------------------------
# Operators
# --------------------------------------------------------------------------
class Operator:
    """A graph node's payload.  ``label`` is for debugging/pretty printing;
    ``eq_key`` drives common-subexpression elimination (EquivalentNodeMergeRule):
    two nodes with equal eq_key and equal dependencies compute the same thing."""

    @property
    def label(self) -> str:
        return type(self).__name__

    def eq_key(self) -> Tuple:
        return (type(self), id(self))


class TransformerOperator(Operator):
    """Wraps a fitted/stateless Transformer; applies it to its single input."""

    def __init__(self, transformer):
        self.transformer = transformer

    @property
    def label(self):
        return type(self.transformer).__name__

    def eq_key(self):
        return ("T", id(self.transformer))


class EstimatorOperator(Operator):
    """Wraps an Estimator (deps: [data] or [data, labels]); output is the
    fitted transformer (a TransformerExpression)."""

    def __init__(self, estimator):
        self.estimator = estimator

    @property
    def label(self):
        return "fit:" + type(self.estimator).__name__

    def eq_key(self):
        return ("E", id(self.estimator))


class DelegatingOperator(Operator):
    """Applies the transformer produced by dep[0] to the data at dep[1].
    This is how ``pipeline.and_then(est, data)`` splices fitted models."""

    @property
    def label(self):
        return "apply"

    def eq_key(self):
        return ("D",)


class DatasetOperator(Operator):
    """A distributed dataset bound directly into the graph (fit data)."""

    def __init__(self, dataset):
        self.dataset = dataset

    @property
    def label(self):
        return "dataset"

    def eq_key(self):
        return ("DS", id(self.dataset))


class DatumOperator(Operator):
    """A single datum bound into the graph."""

    def __init__(self, datum):
        self.datum = datum

    @property
    def label(self):
        return "datum"

    def eq_key(self):
        return ("DT", id(self.datum))


# --------------------------------------------------------------------------
# Graph
# --------------------------------------------------------------------------
@dataclass
class Graph:
    sources: List[SourceId] = field(default_factory=list)
    operators: Dict[NodeId, Operator] = field(default_factory=dict)
    dependencies: Dict[NodeId, List[GraphId]] = field(default_factory=dict)
    sinks: Dict[SinkId, GraphId] = field(default_factory=dict)

    # ------------------------------------------------------------- build
    def copy(self) -> "Graph":
        return Graph(list(self.sources), dict(self.operators),
                     {k: list(v) for k, v in self.dependencies.items()},
                     dict(self.sinks))

    def add_node(self, op: Operator, deps: List[GraphId]) -> NodeId:
        nid = fresh_node_id()
        self.operators[nid] = op
        self.dependencies[nid] = list(deps)
        return nid

    def add_source(self) -> SourceId:
        sid = fresh_source_id()
        self.sources.append(sid)
        return sid

    def add_sink(self, dep: GraphId) -> SinkId:
        kid = fresh_sink_id()
        self.sinks[kid] = dep
        return kid

    def remove_sink(self, kid: SinkId) -> None:
        del self.sinks[kid]

    def replace_dependency(self, old: GraphId, new: GraphId) -> None:
        for nid, deps in self.dependencies.items():
            self.dependencies[nid] = [new if d == old else d for d in deps]
        for kid, d in list(self.sinks.items()):
            if d == old:
                self.sinks[kid] = new

    def replace_source_with_node(self, source: SourceId, op: Operator) -> NodeId:
        """Bind data into a source: the source becomes an operator node."""
        nid = self.add_node(op, [])
        self.replace_dependency(source, nid)
        self.sources.remove(source)
        return nid

    def connect_graph(self, other: "Graph") -> Dict[GraphId, GraphId]:
        """Union another graph into this one (ids are globally unique, so
        this is a plain merge); returns the identity mapping for clarity."""
        self.sources.extend(other.sources)
        self.operators.update(other.operators)
        self.dependencies.update({k: list(v) for k, v in other.dependencies.items()})
        # note: sinks of `other` are NOT merged — the caller wires endpoints.
        return {}

    # ------------------------------------------------------------ queries
    def ancestors(self, gid: GraphId) -> List[GraphId]:
        """All transitive dependencies of gid (excluding gid), post-order."""
        seen, order = set(), []

        def visit(g):
            if g in seen:
                return
            seen.add(g)
            for d in self.dependencies.get(g, []):
                visit(d)
                if d not in order:
                    order.append(d)

        visit(gid)
        return order

    def reachable_from_sinks(self) -> set:
        live = set()
        stack = list(self.sinks.values())
        while stack:
            g = stack.pop()
            if g in live:
                continue
            live.add(g)
            stack.extend(self.dependencies.get(g, []))
        return live

    def consumers(self, gid: GraphId) -> List[NodeId]:
        return [nid for nid, deps in self.dependencies.items() if gid in deps]

    def topo_order(self) -> List[NodeId]:
        """Topological order over operator nodes (reachable from sinks when
        sinks exist; all nodes for unbound prefix pipelines)."""
        live = self.reachable_from_sinks() if self.sinks else set(self.operators)
        seen, order = set(), []

        def visit(g):
            if g in seen or not isinstance(g, NodeId):
                return
            seen.add(g)
            for d in self.dependencies.get(g, []):
                visit(d)
            order.append(g)

        for g in sorted(live, key=lambda x: x.idx):
            visit(g)
        return order

    def pretty(self) -> str:
        lines = []
        for nid in self.topo_order():
            deps = ", ".join(repr(d) for d in self.dependencies[nid])
            lines.append(f"{nid!r}: {self.operators[nid].label} <- [{deps}]")
        for kid, d in self.sinks.items():
            lines.append(f"{kid!r} -> {d!r}")
        return "\n".join(lines)
