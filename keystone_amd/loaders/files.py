"""On-disk dataset loaders.

ref: loaders.CsvDataLoader, CifarLoader (binary format),
NewsgroupsDataLoader (dir-per-class text), AmazonReviewsDataLoader (JSON
reviews), TimitFeaturesDataLoader, ImageNetLoader, LabeledData.

Each loader shards files/rows across ranks deterministically.  Note: this
image (no network) ships no JPEG decoder, so ImageNetLoader accepts .npy /
.pt / PGM(P5) images; synthetic.py is the benchmark path.
"""

from __future__ import annotations

import csv
import json
import os
from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch

from ..parallel.dataset import ItemDataset, TensorDataset, from_items, from_rows
from ..parallel.runtime import get_runtime


@dataclass
class LabeledData:
    """(data, labels) pair of datasets (ref: LabeledData wrapper)."""
    data: object
    labels: TensorDataset


def _shard(seq, rt):
    lo, hi = rt.shard_bounds(len(seq))
    return seq[lo:hi]


def CsvDataLoader(path: str, label_col: int = 0, delimiter: str = ",",
                  runtime=None) -> LabeledData:
    rt = runtime or get_runtime()
    with open(path) as f:
        rows = list(csv.reader(f, delimiter=delimiter))
    rows = _shard(rows, rt)
    labels = torch.tensor([float(r[label_col]) for r in rows])
    feats = torch.tensor([[float(v) for j, v in enumerate(r) if j != label_col]
                          for r in rows], dtype=torch.float32)
    return LabeledData(from_rows(feats.to(rt.device), rt),
                       from_rows(labels.to(rt.device), rt))


def CifarLoader(path: str, runtime=None) -> LabeledData:
    """CIFAR-10 binary format: records of 1 label byte + 3072 pixel bytes
    (ref: CifarLoader)."""
    rt = runtime or get_runtime()
    raw = torch.frombuffer(bytearray(open(path, "rb").read()), dtype=torch.uint8)
    rec = 3073
    n = raw.numel() // rec
    raw = raw[: n * rec].reshape(n, rec)
    lo, hi = rt.shard_bounds(n)
    labels = raw[lo:hi, 0].long()
    imgs = raw[lo:hi, 1:].reshape(-1, 3, 32, 32).float() / 255.0
    return LabeledData(from_rows(imgs.to(rt.device), rt),
                       from_rows(labels.to(rt.device), rt))


def NewsgroupsDataLoader(path: str, runtime=None) -> Tuple[ItemDataset, TensorDataset, List[str]]:
    """Directory-per-class text files (ref: NewsgroupsDataLoader).
    Returns (docs, int labels, class names)."""
    rt = runtime or get_runtime()
    classes = sorted(d for d in os.listdir(path)
                     if os.path.isdir(os.path.join(path, d)))
    files = []
    for ci, c in enumerate(classes):
        cdir = os.path.join(path, c)
        for f in sorted(os.listdir(cdir)):
            files.append((os.path.join(cdir, f), ci))
    files = _shard(files, rt)
    docs, labels = [], []
    for fp, ci in files:
        with open(fp, errors="replace") as fh:
            docs.append(fh.read())
        labels.append(ci)
    return (from_items(docs, rt),
            from_rows(torch.tensor(labels, dtype=torch.long), rt), classes)


def AmazonReviewsDataLoader(path: str, threshold: float = 3.5, runtime=None
                            ) -> Tuple[ItemDataset, TensorDataset]:
    """JSON-lines reviews {reviewText, overall} -> (text, binary label)
    (ref: AmazonReviewsDataLoader)."""
    rt = runtime or get_runtime()
    with open(path) as f:
        lines = f.readlines()
    lines = _shard(lines, rt)
    docs, labels = [], []
    for ln in lines:
        try:
            rec = json.loads(ln)
        except json.JSONDecodeError:
            continue
        docs.append(rec.get("reviewText", ""))
        labels.append(1 if float(rec.get("overall", 0)) > threshold else 0)
    return (from_items(docs, rt),
            from_rows(torch.tensor(labels, dtype=torch.long), rt))


def TimitFeaturesDataLoader(feature_path: str, label_path: str, runtime=None
                            ) -> LabeledData:
    """Precomputed MFCC frames (.npy/.pt or whitespace text) + labels
    (ref: TimitFeaturesDataLoader)."""
    rt = runtime or get_runtime()
    feats = _load_matrix(feature_path)
    labels = _load_matrix(label_path).reshape(-1).long()
    lo, hi = rt.shard_bounds(feats.shape[0])
    return LabeledData(from_rows(feats[lo:hi].float().to(rt.device), rt),
                       from_rows(labels[lo:hi].to(rt.device), rt))


def _load_matrix(path: str) -> torch.Tensor:
    if path.endswith(".pt"):
        return torch.load(path)
    if path.endswith(".npy"):
        import numpy as np
        return torch.from_numpy(np.load(path))
    import numpy as np
    return torch.from_numpy(np.loadtxt(path))


def _read_pgm(path: str) -> torch.Tensor:
    with open(path, "rb") as f:
        data = f.read()
    assert data[:2] == b"P5", "only binary PGM (P5) supported"
    parts = data.split(maxsplit=4)
    w, h, maxv = int(parts[1]), int(parts[2]), int(parts[3])
    img = torch.frombuffer(bytearray(parts[4][: w * h]), dtype=torch.uint8)
    return img.reshape(h, w).float() / maxv


_PIL_EXT = (".jpg", ".jpeg", ".png", ".ppm", ".bmp")


def _decode_file(fp: str, color: bool, device) -> Optional[torch.Tensor]:
    if fp.endswith(".pgm"):
        return _read_pgm(fp).to(device)
    if fp.endswith(".npy"):
        import numpy as np
        return torch.from_numpy(np.load(fp)).float().to(device)
    if fp.endswith(".pt"):
        return torch.load(fp).float().to(device)
    if fp.lower().endswith(_PIL_EXT):
        from .input_pipeline import decode_image_bytes
        with open(fp, "rb") as fh:
            return decode_image_bytes(fh.read(), color=color).to(device)
    return None


def ImageNetLoader(path: str, label_map: Optional[dict] = None, runtime=None,
                   color: bool = True) -> Tuple[ItemDataset, TensorDataset]:
    """Real-format image ingestion (ref: ImageNetLoader + ImageLoaderUtils).

    Two layouts:
      * directory-per-class of image files — JPEG/PNG/PPM/BMP (PIL
        decode), plus .pgm/.npy/.pt;
      * a directory of ``*.tar`` shards whose members are named
        ``<class>/<image>.JPEG`` — the reference's tar-of-JPEG sharding.
        Tar files are sharded across ranks when there are at least
        world_size of them (each rank streams whole shards); otherwise
        members are sharded individually.
    Labels come from the class directory/prefix via ``label_map`` or the
    sorted-name index."""
    rt = runtime or get_runtime()
    tars = sorted(f for f in os.listdir(path) if f.endswith(".tar")) \
        if os.path.isdir(path) else []
    imgs, labels = [], []
    if tars:
        import tarfile
        from .input_pipeline import decode_image_bytes
        # discover classes across all shards for a stable label index
        if label_map is None:
            names = set()
            for t in tars:
                with tarfile.open(os.path.join(path, t)) as tf:
                    for m in tf.getmembers():
                        if "/" in m.name:
                            names.add(m.name.split("/", 1)[0])
            label_map = {c: i for i, c in enumerate(sorted(names))}
        if len(tars) >= rt.world_size:
            my_tars, my_members = _shard(tars, rt), None
        else:
            my_tars, my_members = tars, True
        idx = 0
        for t in my_tars:
            with tarfile.open(os.path.join(path, t)) as tf:
                for m in sorted(tf.getmembers(), key=lambda m: m.name):
                    if not m.isfile() or "/" not in m.name:
                        continue
                    cls = m.name.split("/", 1)[0]
                    if cls not in label_map:
                        continue
                    if my_members is not None \
                            and idx % rt.world_size != rt.rank:
                        idx += 1
                        continue
                    idx += 1
                    data = tf.extractfile(m).read()
                    imgs.append(decode_image_bytes(data, color).to(rt.device))
                    labels.append(label_map[cls])
        return (from_items(imgs, rt),
                from_rows(torch.tensor(labels, dtype=torch.long), rt))

    entries = []
    classes = sorted(d for d in os.listdir(path)
                     if os.path.isdir(os.path.join(path, d)))
    cmap = label_map or {c: i for i, c in enumerate(classes)}
    for c in classes:
        for f in sorted(os.listdir(os.path.join(path, c))):
            entries.append((os.path.join(path, c, f), cmap[c]))
    entries = _shard(entries, rt)
    for fp, ci in entries:
        t = _decode_file(fp, color, rt.device)
        if t is None:
            continue
        imgs.append(t)
        labels.append(ci)
    return (from_items(imgs, rt),
            from_rows(torch.tensor(labels, dtype=torch.long), rt))


def VOCLoader(path: str, num_classes: int = 20, runtime=None):
    """VOC-style multi-label: a JSON index {file: [class ids]} next to the
    images (ref: VOCLoader; the XML-annotation variant reduces to this)."""
    rt = runtime or get_runtime()
    with open(os.path.join(path, "index.json")) as f:
        index = sorted(json.load(f).items())
    index = _shard(index, rt)
    imgs, labels = [], []
    for fname, cls_list in index:
        fp = os.path.join(path, fname)
        if fp.endswith(".pgm"):
            imgs.append(_read_pgm(fp).to(rt.device))
        elif fp.endswith(".npy"):
            import numpy as np
            imgs.append(torch.from_numpy(np.load(fp)).float().to(rt.device))
        else:
            continue
        y = torch.zeros(num_classes)
        y[torch.tensor(cls_list, dtype=torch.long)] = 1.0
        labels.append(y)
    return (from_items(imgs, rt),
            from_rows(torch.stack(labels).to(rt.device), rt))


class ImageExtractor:
    """LabeledData -> images (ref: nodes.images.ImageExtractor)."""

    def __call__(self, ld: LabeledData):
        return ld.data


class LabelExtractor:
    """LabeledData -> labels (ref: nodes.images.LabelExtractor)."""

    def __call__(self, ld: LabeledData):
        return ld.labels
