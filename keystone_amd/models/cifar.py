"""CIFAR random-patch convolution pipeline.

ref: pipelines/images/cifar/RandomPatchCifar —
random patches -> ZCA whitening -> Convolver(whitened patch bank) ->
SymmetricRectifier -> Pooler -> vectorize -> least squares -> argmax.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

from ..ops.images import (Convolver, ImageVectorizer, Pooler,
                          SymmetricRectifier)
from ..ops.learning import LinearMapEstimator
from ..ops.learning.whitening import ZCAWhitenerEstimator
from ..ops.util import ClassLabelIndicatorsFromIntLabels, MaxClassifier
from ..parallel.dataset import TensorDataset
from ..workflow.pipeline import Pipeline


@dataclass
class CifarConfig:
    num_filters: int = 256
    patch_size: int = 6
    patch_sample: int = 10000
    pool_size: int = 13
    pool_stride: int = 13
    alpha: float = 0.25
    lam: float = 10.0
    num_classes: int = 10
    zca_eps: float = 0.1
    seed: int = 0


def fit_cifar_patches(train_imgs: TensorDataset, labels_int: TensorDataset,
                      conf: CifarConfig = CifarConfig()) -> Pipeline:
    """train_imgs: [B, C, H, W] shard."""
    X = train_imgs.data
    rt = train_imgs.runtime
    B, C, H, W = X.shape
    g = torch.Generator().manual_seed(conf.seed + rt.rank)
    # sample random patches from the local shard
    n_local = max(1, conf.patch_sample // max(1, rt.world_size))
    bi = torch.randint(0, B, (n_local,), generator=g)
    yi = torch.randint(0, H - conf.patch_size + 1, (n_local,), generator=g)
    xi = torch.randint(0, W - conf.patch_size + 1, (n_local,), generator=g)
    patches = torch.stack([
        X[b, :, y:y + conf.patch_size, x:x + conf.patch_size].reshape(-1)
        for b, y, x in zip(bi, yi, xi)]).to(rt.device)
    zca = ZCAWhitenerEstimator(conf.zca_eps).fit(TensorDataset(patches, rt))
    white = zca.apply_batch(patches)
    # deterministic filter choice: first num_filters whitened patches of
    # rank 0's stream (replicated via the gathered sample)
    allw = TensorDataset(white, rt).collect()
    filt = allw[: conf.num_filters].reshape(conf.num_filters, C,
                                            conf.patch_size, conf.patch_size)
    filt = filt / filt.reshape(conf.num_filters, -1).norm(dim=1).clamp_min(
        1e-8).reshape(-1, 1, 1, 1)

    featurize = (Convolver(filt, mode="im2col")
                 .and_then(SymmetricRectifier(conf.alpha))
                 .and_then(Pooler(conf.pool_stride, conf.pool_size, "sum"))
                 .and_then(ImageVectorizer()))
    ind = ClassLabelIndicatorsFromIntLabels(conf.num_classes)
    labels = TensorDataset(ind.apply_batch(labels_int.data).float(), rt)
    return (featurize
            .and_then(LinearMapEstimator(conf.lam, fit_intercept=True),
                      train_imgs, labels)
            .and_then(MaxClassifier()))
