"""VOC 2007 SIFT Fisher-Vector pipeline (multi-label, mAP-evaluated).

ref: pipelines/images/voc/VOCSIFTFisher — SIFT -> PCA ->
GMM -> FisherVector -> normalization -> BlockLeastSquares (multi-label
+-1 targets) -> per-class scores -> MeanAveragePrecisionEvaluator.
"""

from __future__ import annotations

from dataclasses import dataclass


from ..ops.images import FisherVector, GrayScaler, SIFTExtractor
from ..ops.learning import BlockLeastSquaresEstimator, DistributedPCAEstimator
from ..ops.learning.gmm import GaussianMixtureModelEstimator
from ..ops.stats import ColumnSampler, NormalizeRows, SignedHellingerMapper
from ..parallel.dataset import TensorDataset
from ..workflow.pipeline import Pipeline


@dataclass
class VOCConfig:
    num_classes: int = 20
    sift_step: int = 6
    sift_bin: int = 6
    sift_scales: int = 2
    pca_dims: int = 64
    gmm_k: int = 16
    n_sample: int = 50000
    block_size: int = 4096
    solver_epochs: int = 2
    lam: float = 1e-4
    seed: int = 0


def fit_voc(train_imgs: TensorDataset, multilabels: TensorDataset,
            conf: VOCConfig = VOCConfig()) -> Pipeline:
    """multilabels: [B, num_classes] in {0,1}; returns a pipeline emitting
    per-class scores (feed to MeanAveragePrecisionEvaluator)."""
    rt = train_imgs.runtime
    sift = GrayScaler().and_then(
        SIFTExtractor(conf.sift_step, conf.sift_bin, conf.sift_scales))
    desc = sift.apply(train_imgs).get()
    sample = ColumnSampler(conf.n_sample, seed=conf.seed).apply_dataset(desc)
    pca = DistributedPCAEstimator(conf.pca_dims).fit(sample)
    proj = TensorDataset(pca.apply_batch(sample.data), rt)
    gmm = GaussianMixtureModelEstimator(conf.gmm_k, max_iterations=20,
                                        seed=conf.seed).fit(proj)
    fv = FisherVector(gmm)
    post = SignedHellingerMapper().and_then(NormalizeRows())
    feats = post.apply(TensorDataset(
        fv.apply_batch(pca.apply_batch(desc.data)), rt)).get()

    Y = multilabels.data.float() * 2 - 1   # {0,1} -> ±1 indicators
    solver = BlockLeastSquaresEstimator(conf.block_size, conf.solver_epochs,
                                        conf.lam)
    model = solver.fit(feats, TensorDataset(Y, rt))
    return sift.and_then(pca).and_then(fv).and_then(post).and_then(model)
