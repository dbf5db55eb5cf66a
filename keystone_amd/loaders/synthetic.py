"""Synthetic dataset generators for the five benchmark configs.

BASELINE.json mandates synthetic data / random-init weights (no network
for datasets), so every driver config has a generator here producing the
same shapes as the real loaders:

  1. 20-Newsgroups   -> random documents over a Zipf vocabulary, 20 classes
  2. MNIST           -> [n, 784] fp32 rows, 10 classes
  3. TIMIT           -> [n, 440] MFCC+context frames, 147 phone classes
  4. Amazon reviews  -> longer documents, binary labels
  5. ImageNet        -> 256x256 grayscale-ish images, 1000 classes

Sharding: documents are per-index deterministic (identical global corpus
at any world size); the dense/image generators draw independent per-rank
streams (weak-scaling benchmarks fix per-rank work, not the global set).
"""

from __future__ import annotations

from typing import List, Tuple

import torch

from ..parallel.dataset import ItemDataset, TensorDataset, from_items, from_rows
from ..parallel.runtime import get_runtime

_WORDS = None


def _vocab(size: int) -> List[str]:
    global _WORDS
    if _WORDS is None or len(_WORDS) < size:
        _WORDS = [f"w{i:05d}" for i in range(size)]
    return _WORDS[:size]


def synthetic_documents(n_docs: int, num_classes: int = 20,
                        vocab_size: int = 5000, doc_len: int = 80,
                        seed: int = 0, runtime=None
                        ) -> Tuple[ItemDataset, TensorDataset]:
    """Documents with class-dependent Zipf word distributions (so linear
    models genuinely separate them) + int labels."""
    rt = runtime or get_runtime()
    lo, hi = rt.shard_bounds(n_docs)
    words = _vocab(vocab_size)
    # Zipf base distribution + per-class boosted sub-vocabulary
    ranks = torch.arange(1, vocab_size + 1, dtype=torch.float64)
    base = 1.0 / ranks
    docs, labels = [], []
    per_class = max(10, vocab_size // (4 * num_classes))
    for i in range(lo, hi):
        g = torch.Generator().manual_seed(seed * 1_000_003 + i)
        c = int(torch.randint(0, num_classes, (1,), generator=g).item())
        probs = base.clone()
        probs[c * per_class % vocab_size:(c * per_class % vocab_size) + per_class] *= 40.0
        probs /= probs.sum()
        idx = torch.multinomial(probs, doc_len, replacement=True, generator=g)
        docs.append(" ".join(words[j] for j in idx))
        labels.append(c)
    return (from_items(docs, rt),
            from_rows(torch.tensor(labels, dtype=torch.long), rt))


def synthetic_dense(n: int, d: int, num_classes: int, seed: int = 0,
                    runtime=None, noise: float = 0.5,
                    device=None) -> Tuple[TensorDataset, TensorDataset]:
    """Gaussian class-cluster rows (MNIST/TIMIT shapes): class means drawn
    once, rows = mean + noise."""
    rt = runtime or get_runtime()
    dev = device or rt.device
    g0 = torch.Generator().manual_seed(seed)
    means = torch.randn(num_classes, d, generator=g0)
    lo, hi = rt.shard_bounds(n)
    n_local = hi - lo
    g = torch.Generator().manual_seed(seed + 17 + rt.rank)
    labels = torch.randint(0, num_classes, (n_local,), generator=g)
    X = means[labels] + noise * torch.randn(n_local, d, generator=g)
    return (from_rows(X.to(dev), rt), from_rows(labels.to(dev), rt))


def synthetic_mnist(n: int = 60000, seed: int = 0, runtime=None, device=None):
    return synthetic_dense(n, 784, 10, seed, runtime, device=device)


def synthetic_timit(n: int = 100000, context: int = 20, seed: int = 0,
                    runtime=None, device=None):
    """TIMIT MFCC frames with +-context splicing: 440 dims, 147 classes
    (ref: TimitFeaturesDataLoader shapes)."""
    d = 440
    return synthetic_dense(n, d, 147, seed, runtime, device=device)


def synthetic_images(n: int, h: int = 256, w: int = 256, num_classes: int = 1000,
                     seed: int = 0, runtime=None, device=None,
                     channels: int = 1, class_structure: bool = False,
                     pattern_seed: int = 1234):
    """ImageNet-shaped synthetic images in [0,1]: smooth random blobs
    (low-frequency noise) so SIFT-like featurizers see structure.

    ``class_structure=True`` additionally stamps each image with its
    class's fixed random texture patch at jittered positions — labels
    become *visually learnable*, so the FV pipeline's test-set accuracy
    measures real generalization (used by the statistical-parity tests;
    the throughput bench keeps random labels).

    Returns ([n_local, h, w] gray or [n_local, C, h, w] color
    TensorDataset, int labels)."""
    rt = runtime or get_runtime()
    dev = device or rt.device
    lo, hi = rt.shard_bounds(n)
    n_local = hi - lo
    g = torch.Generator().manual_seed(seed + 23 + rt.rank)
    # low-res noise upsampled = smooth structure, plus fine detail
    coarse = torch.rand(n_local, channels, h // 8, w // 8, generator=g)
    imgs = torch.nn.functional.interpolate(coarse, size=(h, w),
                                           mode="bilinear",
                                           align_corners=False)
    imgs = 0.8 * imgs + 0.2 * torch.rand(n_local, channels, h, w, generator=g)
    labels = torch.randint(0, num_classes, (n_local,), generator=g)
    if class_structure:
        ps = max(8, h // 8)
        # class patterns are tied to pattern_seed, NOT the data seed, so
        # train/test splits drawn with different seeds share the classes
        gc = torch.Generator().manual_seed(pattern_seed)
        patterns = torch.rand(num_classes, channels, ps, ps, generator=gc)
        n_stamps = max(3, (h // ps))
        for i in range(n_local):
            pat = patterns[labels[i]]
            for _ in range(n_stamps):
                y0 = int(torch.randint(0, h - ps + 1, (1,), generator=g))
                x0 = int(torch.randint(0, w - ps + 1, (1,), generator=g))
                imgs[i, :, y0:y0 + ps, x0:x0 + ps] = \
                    0.35 * imgs[i, :, y0:y0 + ps, x0:x0 + ps] + 0.65 * pat
    if channels == 1:
        imgs = imgs.squeeze(1)
    return (from_rows(imgs.to(dev), rt), from_rows(labels.to(dev), rt))
