"""Multi-process CPU distributed test harness.

The reference tested multi-worker semantics with a local[2..4] SparkContext
(LocalSparkContext fixture); here we spawn N OS processes with a gloo
process group over a FileStore (no ports, no hostname resolution) — the
exact code paths that run over RCCL on the GPU box run here on CPU.
"""

from __future__ import annotations

import datetime
import os
import tempfile
import traceback

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _to_plain(x):
    """Detach results from torch's shared-memory IPC: tensors are sent by
    value (the child exits before the parent could detach a shared FD)."""
    if isinstance(x, torch.Tensor):
        return ("__tensor__", x.numpy().copy())
    if isinstance(x, (list, tuple)):
        return type(x)(_to_plain(v) for v in x)
    if isinstance(x, dict):
        return {k: _to_plain(v) for k, v in x.items()}
    return x


def _from_plain(x):
    if isinstance(x, tuple) and len(x) == 2 and x[0] == "__tensor__":
        return torch.from_numpy(x[1])
    if isinstance(x, (list, tuple)):
        return type(x)(_from_plain(v) for v in x)
    if isinstance(x, dict):
        return {k: _from_plain(v) for k, v in x.items()}
    return x


def _worker(rank: int, world: int, store_path: str, fn, args, q):
    try:
        store = dist.FileStore(store_path, world)
        dist.init_process_group("gloo", store=store, rank=rank,
                                world_size=world,
                                timeout=datetime.timedelta(seconds=120))
        from keystone_amd.parallel.runtime import Runtime, set_runtime
        set_runtime(Runtime(rank=rank, world_size=world,
                            device=torch.device("cpu"), backend="gloo"))
        result = fn(rank, world, *args)
        q.put((rank, "ok", _to_plain(result)))
    except Exception:
        q.put((rank, "err", traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def run_distributed(fn, world_size: int = 2, args: tuple = ()):
    """Run ``fn(rank, world, *args)`` in ``world_size`` processes; returns
    the list of per-rank results (rank order). Raises on any rank error."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    with tempfile.TemporaryDirectory() as td:
        store_path = os.path.join(td, "store")
        procs = [ctx.Process(target=_worker,
                             args=(r, world_size, store_path, fn, args, q))
                 for r in range(world_size)]
        for p in procs:
            p.start()
        results = {}
        for _ in range(world_size):
            rank, status, payload = q.get()
            if status == "err":
                for p in procs:
                    p.terminate()
                raise RuntimeError(f"rank {rank} failed:\n{payload}")
            results[rank] = _from_plain(payload)
        for p in procs:
            p.join(timeout=60)
    return [results[r] for r in range(world_size)]
