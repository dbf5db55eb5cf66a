"""hipGraph-captured pipeline serving.

The transform (serving) path of a fitted pipeline is a fixed chain of
kernel launches over a fixed-shape shard — exactly the launch-bound inner
loop hipGraphs exist for.  ``capture_apply`` records ONE execution of the
whole pipeline (library GEMMs, MIOpen convs and our ctypes-launched HIP
kernels alike — stream capture sees everything on torch's current stream)
and replays it per batch, collapsing hundreds of launch round-trips into
one ``hipGraphLaunch``.

Constraints (checked/owned by the caller):
  * fixed input shape/dtype/device (the captured shard shape);
  * transform path only — no estimator fits, no collectives, no host
    syncs inside (true for every Transformer in this framework);
  * inputs are copied into a static buffer before replay and outputs
    cloned out after.
"""

from __future__ import annotations


import torch

from ..parallel.dataset import TensorDataset
from ..parallel.runtime import get_runtime

__all__ = ["capture_apply", "CapturedApply"]


class CapturedApply:
    def __init__(self, pipeline, example: torch.Tensor):
        assert example.is_cuda, "graph capture needs a CUDA shard"
        rt = get_runtime()
        self.pipeline = pipeline
        self.static_in = example.clone()

        def run() -> torch.Tensor:
            out = pipeline.apply(TensorDataset(self.static_in, rt)).get()
            return out.data if isinstance(out, TensorDataset) else out

        # warm up on a side stream (allocator + library workspaces settle)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                run()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_out = run()

    def __call__(self, X: torch.Tensor) -> torch.Tensor:
        if X.shape != self.static_in.shape:
            raise ValueError(
                f"captured for shape {tuple(self.static_in.shape)}, "
                f"got {tuple(X.shape)}")
        self.static_in.copy_(X)
        self.graph.replay()
        return self.static_out

    def apply_batch(self, X: torch.Tensor) -> torch.Tensor:
        return self(X).clone()


def capture_apply(pipeline, example: torch.Tensor) -> CapturedApply:
    """Capture ``pipeline.apply`` on ``example``'s shape into a hipGraph."""
    return CapturedApply(pipeline, example)
