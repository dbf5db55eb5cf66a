"""Real-format ingestion: JPEG decode, tar-of-JPEG shards,
and the pinned-H2D batch pipeline (CPU mode here; H2D path is exercised by
the GPU bench)."""

import io
import os
import tarfile

import pytest
import torch

PIL = pytest.importorskip("PIL")
from PIL import Image  # noqa: E402

from keystone_amd.loaders.files import ImageNetLoader
from keystone_amd.loaders.input_pipeline import (ImageBatchPipeline,
                                                 decode_image_bytes,
                                                 encode_jpeg_batch)


def _jpeg_bytes(h=32, w=32, seed=0):
    g = torch.Generator().manual_seed(seed)
    arr = (torch.rand(h, w, 3, generator=g) * 255).to(torch.uint8).numpy()
    buf = io.BytesIO()
    Image.fromarray(arr).save(buf, format="JPEG", quality=95)
    return buf.getvalue()


def test_decode_image_bytes_roundtrip():
    data = _jpeg_bytes()
    t = decode_image_bytes(data, color=True)
    assert t.shape == (3, 32, 32)
    assert 0.0 <= t.min() and t.max() <= 1.0
    g = decode_image_bytes(data, color=False)
    assert g.shape == (32, 32)


def _smooth_imgs(n, h, w):
    # JPEG keeps smooth content; random noise is its worst case
    y = torch.linspace(0, 1, h).view(1, 1, h, 1)
    x = torch.linspace(0, 1, w).view(1, 1, 1, w)
    phase = torch.arange(n).view(n, 1, 1, 1) / max(1, n)
    c = torch.arange(3).view(1, 3, 1, 1) / 3.0
    return (0.5 + 0.5 * torch.sin(3 * (x + y + phase + c))).clamp(0, 1)


def test_encode_decode_batch_close():
    imgs = _smooth_imgs(4, 24, 24)
    blobs = encode_jpeg_batch(imgs, quality=95, workers=2)
    back = torch.stack([decode_image_bytes(b) for b in blobs])
    assert back.shape == imgs.shape
    assert (back - imgs).abs().mean() < 0.05  # JPEG is lossy


def test_imagenet_loader_jpeg_dirs(tmp_path):
    for ci, cls in enumerate(["n01", "n02"]):
        d = tmp_path / cls
        d.mkdir()
        for j in range(3):
            (d / f"img{j}.jpg").write_bytes(_jpeg_bytes(seed=ci * 10 + j))
    imgs, labels = ImageNetLoader(str(tmp_path))
    assert labels.data.tolist() == [0, 0, 0, 1, 1, 1]
    assert all(x.shape == (3, 32, 32) for x in imgs.items())


def test_imagenet_loader_tar_shards(tmp_path):
    for shard in range(2):
        with tarfile.open(tmp_path / f"shard{shard}.tar", "w") as tf:
            for ci, cls in enumerate(["n01", "n02"]):
                data = _jpeg_bytes(seed=shard * 100 + ci)
                info = tarfile.TarInfo(f"{cls}/im{shard}.JPEG")
                info.size = len(data)
                tf.addfile(info, io.BytesIO(data))
    imgs, labels = ImageNetLoader(str(tmp_path))
    assert sorted(labels.data.tolist()) == [0, 0, 1, 1]
    assert all(x.shape == (3, 32, 32) for x in imgs.items())


def test_image_batch_pipeline_cpu():
    imgs = _smooth_imgs(10, 16, 16)
    blobs = encode_jpeg_batch(imgs, workers=2)
    pipe = ImageBatchPipeline(blobs, list(range(10)), batch_size=4,
                              device=None, workers=2)
    batches = list(pipe)
    assert len(batches) == 3
    assert [b.shape[0] for b, _ in batches] == [4, 4, 2]
    labs = torch.cat([l for _, l in batches])
    assert labs.tolist() == list(range(10))
    got = torch.cat([b for b, _ in batches])
    assert (got - imgs).abs().mean() < 0.06


@pytest.mark.gpu
def test_image_batch_pipeline_h2d():
    imgs = _smooth_imgs(12, 32, 32)
    blobs = encode_jpeg_batch(imgs, workers=4)
    dev = torch.device("cuda", 0)
    pipe = ImageBatchPipeline(blobs, list(range(12)), batch_size=4,
                              device=dev, workers=4)
    got = torch.cat([b for b, _ in pipe])
    assert got.is_cuda
    assert (got.cpu() - imgs).abs().mean() < 0.06


def test_process_image_pipeline_cpu():
    from keystone_amd.loaders.input_pipeline import ProcessImagePipeline
    imgs = _smooth_imgs(10, 24, 24)
    blobs = encode_jpeg_batch(imgs, workers=2)
    pipe = ProcessImagePipeline(blobs, list(range(10)), batch_size=4,
                                device=None, workers=2)
    batches = list(pipe)
    assert [b.shape[0] for b, _ in batches] == [4, 4, 2]
    labs = torch.cat([l for _, l in batches])
    assert labs.tolist() == list(range(10))
    got = torch.cat([b for b, _ in batches])
    assert (got - imgs).abs().mean() < 0.06


def _tar_loader_world2(rank, world, tmpdir):
    from keystone_amd.loaders.files import ImageNetLoader
    imgs, labels = ImageNetLoader(tmpdir)
    return labels.data, torch.stack(list(imgs.items())).shape


def test_imagenet_tar_shards_split_across_ranks(tmp_path):
    """4 tar shards, world 2: each rank streams 2 whole shards."""
    from tests.dist_util import run_distributed
    for shard in range(4):
        with tarfile.open(tmp_path / f"s{shard}.tar", "w") as tf:
            for ci, cls in enumerate(["n01", "n02"]):
                data = _jpeg_bytes(seed=shard * 100 + ci)
                info = tarfile.TarInfo(f"{cls}/im{shard}.JPEG")
                info.size = len(data)
                tf.addfile(info, io.BytesIO(data))
    results = run_distributed(_tar_loader_world2, 2, args=(str(tmp_path),))
    all_labels = torch.cat([r[0] for r in results])
    assert sorted(all_labels.tolist()) == [0, 0, 0, 0, 1, 1, 1, 1]
    for _labels, shape in results:
        assert shape == (4, 3, 32, 32)   # 2 shards x 2 images per rank


def test_process_pipeline_dead_worker_aborts_fast():
    """A corrupt blob kills its decode worker; the iterator must raise
    within a few seconds instead of hanging the caller (the short-poll
    liveness check)."""
    import time
    from keystone_amd.loaders.input_pipeline import ProcessImagePipeline
    imgs = _smooth_imgs(8, 24, 24)
    blobs = encode_jpeg_batch(imgs, workers=2)
    blobs[5] = b"not a jpeg at all"
    pipe = ProcessImagePipeline(blobs, list(range(8)), batch_size=4,
                                device=None, workers=2)
    t0 = time.monotonic()
    with pytest.raises(RuntimeError, match="decode worker"):
        list(pipe)
    assert time.monotonic() - t0 < 30
