"""Host -> HBM image input pipeline: decode + pinned staging + async H2D.

ref: ImageLoaderUtils — the reference decoded JPEG on the
JVM per record inside Spark tasks.  MI355X rebuild: a thread pool decodes
(PIL releases the GIL in its C decoders), decoded images are packed into
PINNED host buffers, and each batch crosses PCIe on a dedicated copy
stream so the H2D of batch i+1 overlaps the GPU compute of batch i.

No rocJPEG ships in this ROCm build, so decode is CPU-side; the pipeline
is therefore decode-bound and scales with host cores — the honest
"serving with input" configuration.
"""

from __future__ import annotations

import io
from concurrent.futures import ThreadPoolExecutor
from typing import Iterator, List, Optional, Sequence, Tuple

import torch

__all__ = ["decode_image_bytes", "encode_jpeg_batch", "ImageBatchPipeline",
           "ProcessImagePipeline"]


def decode_image_bytes(data: bytes, color: bool = True) -> torch.Tensor:
    """JPEG/PNG/PPM/... bytes -> float tensor in [0, 1]; [3, H, W] when
    ``color`` else [H, W] luminance."""
    import numpy as np
    from PIL import Image
    img = Image.open(io.BytesIO(data))
    img = img.convert("RGB" if color else "L")
    arr = np.asarray(img)                      # [H, W, 3] or [H, W] uint8
    t = torch.from_numpy(arr.copy())
    if color:
        t = t.permute(2, 0, 1)
    return t.float() / 255.0


def encode_jpeg_batch(imgs: torch.Tensor, quality: int = 85,
                      workers: int = 16) -> List[bytes]:
    """[B, 3, H, W] or [B, H, W] float [0,1] -> list of JPEG bytes
    (threaded; used to synthesize an input corpus when no real dataset is
    available — BASELINE mandates synthetic data)."""
    import numpy as np
    from PIL import Image

    arr8 = (imgs.clamp(0, 1) * 255).to(torch.uint8).cpu()

    def enc(i: int) -> bytes:
        a = arr8[i].numpy()
        if a.ndim == 3:
            a = a.transpose(1, 2, 0)
        buf = io.BytesIO()
        Image.fromarray(a).save(buf, format="JPEG", quality=quality)
        return buf.getvalue()

    with ThreadPoolExecutor(workers) as ex:
        return list(ex.map(enc, range(arr8.shape[0])))


class ImageBatchPipeline:
    """Iterate (gpu_batch [b, ...], labels [b]) from encoded image bytes.

    Double-buffered: while the caller computes on batch i, the pool
    decodes batch i+1 into the other pinned slot and its H2D copy runs on
    the side stream.  ``device=None``/CPU skips pinning and streams.
    """

    def __init__(self, blobs: Sequence[bytes], labels: Sequence[int],
                 batch_size: int = 256, color: bool = True,
                 device: Optional[torch.device] = None, workers: int = 16):
        self.blobs = blobs
        self.labels = torch.as_tensor(labels, dtype=torch.long)
        self.batch = batch_size
        self.color = color
        self.device = device
        self.workers = workers

    def __len__(self) -> int:
        return (len(self.blobs) + self.batch - 1) // self.batch

    def _decode_batch(self, lo: int, hi: int, pool) -> torch.Tensor:
        parts = list(pool.map(
            lambda b: decode_image_bytes(b, self.color),
            self.blobs[lo:hi]))
        return torch.stack(parts)

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        use_gpu = self.device is not None and self.device.type == "cuda"
        pool = ThreadPoolExecutor(self.workers)     # decode fan-out
        stager = ThreadPoolExecutor(1)              # batch coordinator
        copy_stream = torch.cuda.Stream(self.device) if use_gpu else None
        pinned: List[Optional[torch.Tensor]] = [None, None]
        slot_ev: List[Optional[torch.cuda.Event]] = [None, None]
        try:
            n = len(self.blobs)
            ranges = [(lo, min(lo + self.batch, n))
                      for lo in range(0, n, self.batch)]

            def stage(i: int):
                lo, hi = ranges[i]
                cpu = self._decode_batch(lo, hi, pool)
                if not use_gpu:
                    return cpu, self.labels[lo:hi]
                slot = i & 1
                # the slot's previous H2D must be off the wire before the
                # pinned source is overwritten
                if slot_ev[slot] is not None:
                    slot_ev[slot].synchronize()
                if (pinned[slot] is None
                        or pinned[slot].shape[0] < cpu.shape[0]):
                    pinned[slot] = torch.empty_like(cpu).pin_memory()
                buf = pinned[slot][: cpu.shape[0]]
                buf.copy_(cpu)
                with torch.cuda.stream(copy_stream):
                    dev = buf.to(self.device, non_blocking=True)
                ev = torch.cuda.Event()
                ev.record(copy_stream)
                slot_ev[slot] = ev
                return (dev, ev), self.labels[lo:hi]

            # batch i+1 decodes + stages while the caller computes on i
            fut = stager.submit(stage, 0) if ranges else None
            for i in range(len(ranges)):
                data, labs = fut.result()
                fut = stager.submit(stage, i + 1) \
                    if i + 1 < len(ranges) else None
                if use_gpu:
                    dev, ev = data
                    torch.cuda.current_stream(self.device).wait_event(ev)
                    yield dev, labs.to(self.device)
                else:
                    yield data, labs
        finally:
            pool.shutdown(wait=False)
            stager.shutdown(wait=False)


def _mp_decode_worker(worker_id, nworkers, blobs, batch, color, shape,
                      shm_names, done_q, free_q):
    """Decode worker process: owns batches worker_id, worker_id+N, ...;
    writes uint8 frames into its two shared-memory slots."""
    import numpy as np
    from multiprocessing import shared_memory
    shms = [shared_memory.SharedMemory(name=n) for n in shm_names]
    try:
        n = len(blobs)
        nb = (n + batch - 1) // batch
        for bi in range(worker_id, nb, nworkers):
            slot = free_q.get()          # wait until one of OUR slots frees
            lo, hi = bi * batch, min((bi + 1) * batch, n)
            buf = np.ndarray((batch,) + shape, dtype=np.uint8,
                             buffer=shms[slot].buf)
            for j in range(lo, hi):
                from PIL import Image
                img = Image.open(io.BytesIO(blobs[j]))
                img = img.convert("RGB" if color else "L")
                a = np.asarray(img)
                if color:
                    a = a.transpose(2, 0, 1)
                buf[j - lo] = a
            done_q.put((bi, slot, hi - lo))
    finally:
        for s in shms:
            s.close()


class ProcessImagePipeline:
    """Multi-PROCESS decode -> shared-memory uint8 ring -> pinned f32
    staging -> async H2D.  The threaded pipeline tops out ~1.3k img/s on
    256x256 JPEG (the PIL->numpy->stack tail is GIL-bound; more threads
    measured SLOWER); worker processes sidestep the GIL and move frames
    as zero-copy shared-memory uint8.

    Requires uniform image shapes (JPEG corpora of one resolution — the
    tar-shard serving case); ImageBatchPipeline handles the general case.
    """

    def __init__(self, blobs: Sequence[bytes], labels: Sequence[int],
                 batch_size: int = 256, color: bool = True,
                 device: Optional[torch.device] = None, workers: int = 8):
        self.blobs = list(blobs)
        self.labels = torch.as_tensor(labels, dtype=torch.long)
        self.batch = batch_size
        self.color = color
        self.device = device
        self.workers = max(1, workers)
        t0 = decode_image_bytes(self.blobs[0], color)
        self.shape = tuple(t0.shape)       # (3, H, W) or (H, W)

    def __len__(self) -> int:
        return (len(self.blobs) + self.batch - 1) // self.batch

    def __iter__(self):
        import multiprocessing as mp
        import numpy as np
        from multiprocessing import shared_memory
        ctx = mp.get_context("spawn")
        n = len(self.blobs)
        nb = len(self)
        nworkers = min(self.workers, nb) or 1
        frame = int(np.prod(self.shape))
        shms, procs, free_qs = [], [], []
        done_q = ctx.Queue()
        use_gpu = self.device is not None and self.device.type == "cuda"
        copy_stream = torch.cuda.Stream(self.device) if use_gpu else None
        pinned = torch.empty((self.batch,) + self.shape,
                             dtype=torch.float32).pin_memory() \
            if use_gpu else None
        try:
            for w in range(nworkers):
                names = []
                for _ in range(2):
                    shm = shared_memory.SharedMemory(
                        create=True, size=self.batch * frame)
                    shms.append(shm)
                    names.append(shm.name)
                fq = ctx.Queue()
                fq.put(0)
                fq.put(1)
                free_qs.append(fq)
                # each worker only receives ITS batches' blobs
                my = [self.blobs[j] if (j // self.batch) % nworkers == w
                      else b"" for j in range(n)]
                p = ctx.Process(target=_mp_decode_worker,
                                args=(w, nworkers, my, self.batch,
                                      self.color, self.shape, names,
                                      done_q, fq))
                p.daemon = True
                p.start()
                procs.append(p)
            import queue as _queue
            ready = {}
            for want in range(nb):
                # batches must be yielded in order; never block forever
                # on a dead worker (a crashed decode would otherwise hang
                # the caller — and the bench — indefinitely)
                waited = 0.0
                while want not in ready:
                    try:
                        bi, slot, cnt = done_q.get(timeout=2)
                        waited = 0.0
                    except _queue.Empty:
                        # short poll => a crashed decode worker (bad blob,
                        # OOM-killed) is detected in ~2s, not 30
                        if not all(p.is_alive() for p in procs):
                            raise RuntimeError(
                                "ProcessImagePipeline: a decode worker "
                                "died; aborting")
                        waited += 2.0
                        # generous: the FIRST batch includes spawn +
                        # torch import in every worker (minutes on a
                        # cold box); crashes are caught by is_alive
                        if waited >= 300.0:
                            raise RuntimeError(
                                "ProcessImagePipeline: no batch for 300s "
                                "with live workers; aborting")
                        continue
                    ready[bi] = (slot, cnt)
                slot, cnt = ready.pop(want)
                w = want % nworkers
                shm = shms[w * 2 + slot]
                arr = np.ndarray((self.batch,) + self.shape, dtype=np.uint8,
                                 buffer=shm.buf)
                lo = want * self.batch
                labs = self.labels[lo:lo + cnt]
                if use_gpu:
                    src = torch.from_numpy(arr[:cnt])
                    pinned[:cnt].copy_(src.float().div_(255.0))
                    free_qs[w].put(slot)   # shm consumed into pinned
                    with torch.cuda.stream(copy_stream):
                        dev = pinned[:cnt].to(self.device,
                                              non_blocking=True)
                    ev = torch.cuda.Event()
                    ev.record(copy_stream)
                    torch.cuda.current_stream(self.device).wait_event(ev)
                    # the next pinned fill happens on the host AFTER this
                    # H2D: wait for the copy before reusing the buffer
                    ev.synchronize()
                    yield dev, labs.to(self.device)
                else:
                    out = torch.from_numpy(arr[:cnt].copy()).float() / 255.0
                    free_qs[w].put(slot)
                    yield out, labs
        finally:
            for p in procs:
                p.terminate()
            for s in shms:
                s.close()
                try:
                    s.unlink()
                except FileNotFoundError:
                    pass
