"""Serialized-pipeline checkpoint format (save/load of fitted pipelines).

ref: Pipeline.save / Pipeline.load.  KeystoneML Java-
serialized the fitted transformer DAG; that is neither portable nor
inspectable, so the rebuild uses:

  ``<path>/manifest.json``   — format version, DAG topology (nodes, deps,
                               source/out), per-node transformer class path
                               and JSON config;
  ``<path>/node_<k>.safetensors`` — that node's tensor state (fitted
                               weights, means, GMM params, ...), absent for
                               stateless nodes.

Transformers participate via ``get_config()`` / ``get_state()`` /
``from_saved(config, state)`` (see workflow.pipeline.Transformer).  Only
rank 0 writes; loading maps tensors onto the local rank's device.
"""

from __future__ import annotations

import importlib
import json
import os
from typing import Dict

import torch

from ..parallel.runtime import get_runtime
from .graph import Graph, GraphId, TransformerOperator
from .pipeline import Pipeline, Transformer

FORMAT_VERSION = 1

__all__ = ["save_pipeline", "load_pipeline", "FORMAT_VERSION"]


def _class_path(obj) -> str:
    c = type(obj)
    return f"{c.__module__}.{c.__qualname__}"


def _load_class(path: str):
    mod, _, name = path.rpartition(".")
    cls = importlib.import_module(mod)
    for part in name.split("."):
        cls = getattr(cls, part)
    return cls


def save_pipeline(pipeline, path: str) -> None:
    from safetensors.torch import save_file

    g, source, out = pipeline._graph_parts()
    order = g.topo_order()
    for nid in order:
        if not isinstance(g.operators[nid], TransformerOperator):
            raise ValueError(
                "save_pipeline requires a fitted (transformer-only) pipeline; "
                "call .fit() first")

    rt = get_runtime()
    if rt.rank == 0:
        os.makedirs(path, exist_ok=True)
        index = {gid: i for i, gid in enumerate([source] + order)}
        nodes = []
        for nid in order:
            t: Transformer = g.operators[nid].transformer
            entry = {
                "id": index[nid],
                "class": _class_path(t),
                "config": t.get_config(),
                "deps": [index[d] for d in g.dependencies[nid]],
            }
            state = t.get_state()
            if state:
                fname = f"node_{index[nid]}.safetensors"
                save_file({k: v.detach().cpu().contiguous() for k, v in state.items()},
                          os.path.join(path, fname))
                entry["state"] = fname
            nodes.append(entry)
        manifest = {
            "format_version": FORMAT_VERSION,
            "framework": "keystone_amd",
            "torch_version": torch.__version__,
            "source": 0,
            "out": index[out],
            "nodes": nodes,
        }
        with open(os.path.join(path, "manifest.json"), "w") as f:
            json.dump(manifest, f, indent=2)
    rt.barrier()


def load_pipeline(path: str) -> Pipeline:
    from safetensors.torch import load_file

    with open(os.path.join(path, "manifest.json")) as f:
        manifest = json.load(f)
    if manifest["format_version"] > FORMAT_VERSION:
        raise ValueError(f"checkpoint format {manifest['format_version']} is newer "
                         f"than this framework ({FORMAT_VERSION})")

    rt = get_runtime()
    g = Graph()
    src = g.add_source()
    id_map: Dict[int, GraphId] = {manifest["source"]: src}
    for entry in manifest["nodes"]:
        cls = _load_class(entry["class"])
        # the manifest names arbitrary import paths: refuse anything that
        # is not a Transformer (clearer error on corrupt/hostile files
        # than an AttributeError deep in from_saved)
        if not (isinstance(cls, type) and issubclass(cls, Transformer)):
            raise ValueError(f"manifest names non-Transformer class "
                             f"{entry['class']!r}")
        state = {}
        if "state" in entry:
            state = load_file(os.path.join(path, entry["state"]))
            state = {k: v.to(rt.device) for k, v in state.items()}
        t = cls.from_saved(entry["config"], state)
        deps = [id_map[d] for d in entry["deps"]]
        id_map[entry["id"]] = g.add_node(TransformerOperator(t), deps)
    return Pipeline(g, src, id_map[manifest["out"]])
