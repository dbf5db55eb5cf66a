# NOTE: the __main__ guard is MANDATORY — ProcessImagePipeline spawns
# workers that re-import this module; unguarded top-level work would
# re-execute (and re-spawn) in every worker.
import sys, time
sys.path.insert(0, ".")
import torch
from keystone_amd.loaders.input_pipeline import (ImageBatchPipeline,
                                                 ProcessImagePipeline,
                                                 encode_jpeg_batch)


def main():
    imgs = torch.rand(2048, 3, 256, 256).clamp(0, 1)
    blobs = encode_jpeg_batch(imgs, workers=16)
    dev = torch.device("cuda", 0)
    for w in (8, 16):
        pipe = ProcessImagePipeline(blobs, list(range(2048)),
                                    batch_size=256, device=dev, workers=w)
        for b, _ in pipe:
            pass
        t0 = time.perf_counter()
        for b, _ in pipe:
            pass
        torch.cuda.synchronize()
        print(f"proc workers={w}: "
              f"{2048/(time.perf_counter()-t0):.0f} img/s", flush=True)


if __name__ == "__main__":
    main()
