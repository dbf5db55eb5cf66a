"""Distributed runtime for keystone-mi355x.

KeystoneML delegated its cluster runtime to Apache Spark (``SparkContext``,
``RDD[T]``, ``treeReduce``/``treeAggregate``, broadcast, shuffle — ref:
external Spark dependency).  The MI355X rebuild replaces
that layer with an SPMD one-process-per-GPU runtime: every rank holds a
row-shard of each dataset resident in its GPU's 288 GB HBM3E, and the
aggregation patterns Spark expressed as ``treeReduce``/``treeAggregate``
become RCCL collectives over xGMI (``torch.distributed`` backend ``nccl``,
which IS RCCL on ROCm).

Design notes (MI355X-first):
  * xGMI is point-to-point (7 links x ~153 GB/s per GPU) — ring all-reduce is
    bound by a single link, so for the dominant medium-size reductions
    (per-block Gram matrices, ~134 MB fp32 at block=4096) we prefer
    reduce-scatter + all-gather which stripes across links; RCCL picks this
    internally for allreduce at these sizes, so we expose plain
    ``all_reduce`` and tune bucket/block sizes at the call sites.
  * "Broadcast a solved model block" is usually replaced by *replicated
    deterministic local solves*: every rank runs the same Cholesky on the
    same all-reduced Gram — zero communication, bitwise-identical by RCCL
    determinism for a fixed topology.
  * CPU testing: the identical code path runs under the ``gloo`` backend
    with ``world_size`` 2..4 (Spark's ``local[k]`` test fixture analogue,
    ref: LocalSparkContext test trait).
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass, field
from typing import List, Optional

import torch
import torch.distributed as dist

__all__ = ["Runtime", "get_runtime", "init_runtime", "shutdown_runtime"]


@dataclass
class Runtime:
    """Per-process handle on the SPMD job: rank, world size, device, and the
    collective primitives the solver/stat estimators use.

    Replaces Spark's driver/executor split: there is no driver — every rank
    executes the same program, and "driver-side" solves are replicated.
    """

    rank: int = 0
    world_size: int = 1
    device: torch.device = field(default_factory=lambda: torch.device("cpu"))
    backend: Optional[str] = None  # None => single-process, no collectives

    # ---------------------------------------------------------------- info
    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1 and dist.is_initialized()

    @property
    def is_gpu(self) -> bool:
        return self.device.type == "cuda"

    def barrier(self) -> None:
        if self.is_distributed:
            dist.barrier()

    # --------------------------------------------------------- collectives
    # Spark treeReduce(_+_) / treeAggregate  ->  RCCL all-reduce over xGMI.
    # RCCL only moves GPU tensors: small CPU-side metadata tensors (counts,
    # shapes) are bounced through HBM transparently on the nccl backend.
    def _coll_dev(self, t: torch.Tensor) -> torch.Tensor:
        if self.backend == "nccl" and not t.is_cuda:
            return t.to(self.device)
        return t

    def all_reduce_(self, t: torch.Tensor, op: str = "sum") -> torch.Tensor:
        """In-place sum/max/min across ranks. No-op when single-process."""
        if self.is_distributed:
            red = {"sum": dist.ReduceOp.SUM, "max": dist.ReduceOp.MAX,
                   "min": dist.ReduceOp.MIN}[op]
            tc = self._coll_dev(t)
            dist.all_reduce(tc, op=red)
            if tc is not t:
                t.copy_(tc.cpu())
        return t

    def all_reduce_async_(self, t: torch.Tensor):
        """Start an in-place sum all-reduce and return a handle with
        ``.wait()`` (or None when single-process).  RCCL runs the reduction
        on its own stream, so the caller's next GEMM overlaps the wire
        time — used by the block solvers to hide the per-block Gram
        reduction behind the next block's Gram GEMM."""
        if not self.is_distributed:
            return None
        if self.backend == "nccl" and not t.is_cuda:
            self.all_reduce_(t)  # CPU metadata: sync bounce through HBM
            return None
        return dist.all_reduce(t, op=dist.ReduceOp.SUM, async_op=True)

    def all_reduce_many_(self, tensors: List[torch.Tensor], op: str = "sum") -> None:
        """Fused all-reduce of several tensors (one flat buffer => one RCCL
        call; stripes better across the 7 xGMI links than many small calls)."""
        if not self.is_distributed or not tensors:
            return
        if any(t.dtype != tensors[0].dtype for t in tensors):
            # a mixed-dtype cat would silently upcast some summands
            raise TypeError("all_reduce_many_ requires uniform dtype; got "
                            + str([str(t.dtype) for t in tensors]))
        dev = tensors[0].device
        flat = torch.cat([t.reshape(-1).to(dev) for t in tensors])
        self.all_reduce_(flat, op)
        flat = flat.to(dev)
        off = 0
        for t in tensors:
            n = t.numel()
            t.copy_(flat[off:off + n].view_as(t).to(t.device))
            off += n

    def all_gather(self, t: torch.Tensor) -> List[torch.Tensor]:
        if not self.is_distributed:
            return [t]
        tc = self._coll_dev(t.contiguous())
        outs = [torch.empty_like(tc) for _ in range(self.world_size)]
        dist.all_gather(outs, tc)
        if tc is not t and not t.is_cuda:
            outs = [o.cpu() for o in outs]
        return outs

    def all_gather_rows(self, t: torch.Tensor) -> torch.Tensor:
        """Gather row-sharded 2-D tensors with possibly unequal first dims."""
        if not self.is_distributed:
            return t
        n_local = torch.tensor([t.shape[0]], dtype=torch.int64)
        counts = [int(c.item()) for c in self.all_gather(n_local)]
        n_max = max(counts)
        pad = t
        if t.shape[0] < n_max:
            pad = torch.zeros((n_max,) + t.shape[1:], dtype=t.dtype, device=t.device)
            pad[: t.shape[0]] = t
        outs = [torch.empty_like(pad) for _ in range(self.world_size)]
        dist.all_gather(outs, pad.contiguous())
        return torch.cat([o[:c] for o, c in zip(outs, counts)], dim=0)

    def broadcast_(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.is_distributed:
            tc = self._coll_dev(t)
            dist.broadcast(tc, src=src)
            if tc is not t:
                t.copy_(tc.cpu())
        return t

    def broadcast_obj(self, obj, src: int = 0):
        if not self.is_distributed:
            return obj
        box = [obj if self.rank == src else None]
        dist.broadcast_object_list(box, src=src)
        return box[0]

    def all_to_all(self, send: List[torch.Tensor]) -> List[torch.Tensor]:
        """Keyed exchange (Spark shuffle analogue) for row tensors, one
        entry per destination rank; rows may be unequal and zero is fine
        (classes < world_size produce empty sends).

        RCCL path: ONE ``all_to_all_single`` on a flat [sum_rows, row]
        buffer with explicit split sizes — a single fused xGMI kernel that
        handles empty splits, instead of world² point-to-point calls whose
        zero-size sends are backend-dependent.  gloo has no alltoall, so
        the CPU test path exchanges through ``all_gather_object``."""
        if not self.is_distributed:
            return send
        assert len(send) == self.world_size, \
            f"all_to_all wants {self.world_size} entries, got {len(send)}"
        if self.backend == "gloo":
            gathered: List[Optional[list]] = [None] * self.world_size
            dist.all_gather_object(gathered, [t.cpu() for t in send])
            return [gathered[src][self.rank].to(send[0].device)
                    for src in range(self.world_size)]
        t0 = send[0]
        trailing = t0.shape[1:]
        row_elems = 1
        for s in trailing:
            row_elems *= int(s)
        in_splits = [int(s.shape[0]) for s in send]
        splits_t = torch.tensor(in_splits, dtype=torch.int64, device=self.device)
        out_splits_t = torch.empty_like(splits_t)
        dist.all_to_all_single(out_splits_t, splits_t)
        out_splits = [int(x) for x in out_splits_t.cpu()]
        flat_in = self._coll_dev(torch.cat(
            [s.reshape(s.shape[0], row_elems) for s in send]).contiguous())
        flat_out = torch.empty((sum(out_splits), row_elems),
                               dtype=flat_in.dtype, device=flat_in.device)
        dist.all_to_all_single(flat_out, flat_in,
                               output_split_sizes=out_splits,
                               input_split_sizes=in_splits)
        if flat_out.device != t0.device:
            flat_out = flat_out.to(t0.device)
        return [p.reshape((p.shape[0],) + trailing)
                for p in torch.split(flat_out, out_splits, dim=0)]

    # ------------------------------------------------------------- helpers
    def shard_bounds(self, n_total: int) -> tuple:
        """Row range [lo, hi) owned by this rank under balanced sharding."""
        base, rem = divmod(n_total, self.world_size)
        lo = self.rank * base + min(self.rank, rem)
        hi = lo + base + (1 if self.rank < rem else 0)
        return lo, hi

    def synchronize(self) -> None:
        if self.is_gpu:
            torch.cuda.synchronize(self.device)


_RUNTIME: Optional[Runtime] = None


def init_runtime(device: Optional[str] = None, backend: Optional[str] = None,
                 timeout_s: Optional[int] = None) -> Runtime:
    """Initialise the process's runtime from torchrun-style env vars.

    Single-process (no RANK in env): a trivial runtime, no process group.
    Multi-process: ``nccl`` (=RCCL) when CUDA is available, else ``gloo``.
    A hung collective aborts the job after ``timeout_s``
    (KEYSTONE_COLL_TIMEOUT_S, default 600) instead of deadlocking —
    the failure-detection story for single-node SPMD.
    """
    global _RUNTIME
    if _RUNTIME is not None:
        return _RUNTIME
    if timeout_s is None:
        timeout_s = int(os.environ.get("KEYSTONE_COLL_TIMEOUT_S", "600"))

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_gpu = torch.cuda.is_available() if device is None else device.startswith("cuda")
    if use_gpu:
        dev = torch.device("cuda", local_rank % max(1, torch.cuda.device_count()))
        torch.cuda.set_device(dev)
    else:
        dev = torch.device("cpu")

    if world > 1:
        if backend is None:
            backend = "nccl" if use_gpu else "gloo"
        if not dist.is_initialized():
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29571")
            kwargs = {}
            if backend == "nccl" and use_gpu:
                # bind the communicator to this rank's device up front
                # (avoids device-guess hangs on heterogeneous mappings)
                kwargs["device_id"] = dev
            dist.init_process_group(backend=backend, rank=rank, world_size=world,
                                    timeout=datetime.timedelta(seconds=timeout_s),
                                    **kwargs)
    else:
        backend = None

    _RUNTIME = Runtime(rank=rank, world_size=world, device=dev, backend=backend)
    return _RUNTIME


def get_runtime() -> Runtime:
    """The process-wide runtime (lazily created single-process if unset)."""
    global _RUNTIME
    if _RUNTIME is None:
        _RUNTIME = init_runtime()
    return _RUNTIME


def set_runtime(rt: Runtime) -> None:
    """Install an explicit runtime (used by in-process multi-rank tests)."""
    global _RUNTIME
    _RUNTIME = rt


def shutdown_runtime() -> None:
    global _RUNTIME
    if _RUNTIME is not None and dist.is_initialized():
        dist.destroy_process_group()
    _RUNTIME = None
