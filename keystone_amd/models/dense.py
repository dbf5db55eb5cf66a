"""Dense-feature pipelines: MNIST random features and TIMIT.

ref: pipelines/images/mnist/MnistRandomFFT and
pipelines/speech/timit/TimitPipeline — driver configs 2 and 3.

  MNIST: gather_b [RandomSignNode -> PaddedFFT] -> VectorCombiner ->
         (LinearMap | BlockLS) -> MaxClassifier; or the cosine variant.
  TIMIT: StandardScaler -> CosineRandomFeatures(D total) ->
         BlockLeastSquares(blockSize, epochs, lambda) -> MaxClassifier.
"""

from __future__ import annotations

from dataclasses import dataclass

from ..ops.learning import BlockLeastSquaresEstimator
from ..ops.stats import (CosineRandomFeatures, PaddedFFT, RandomSignNode,
                         StandardScaler)
from ..ops.util import (ClassLabelIndicatorsFromIntLabels, MaxClassifier,
                        VectorCombiner)
from ..parallel.dataset import TensorDataset
from ..workflow.pipeline import Pipeline, gather


@dataclass
class MnistConfig:
    input_dim: int = 784
    num_ffts: int = 4
    num_classes: int = 10
    lam: float = 1e-3
    block_size: int = 2048
    num_iter: int = 3
    seed: int = 0


def build_mnist_randomfft(train, labels_int, conf: MnistConfig = MnistConfig()
                          ) -> Pipeline:
    branches = [RandomSignNode(conf.input_dim, seed=conf.seed + i)
                .and_then(PaddedFFT()) for i in range(conf.num_ffts)]
    feats = gather(branches).and_then(VectorCombiner())
    ind = ClassLabelIndicatorsFromIntLabels(conf.num_classes)
    labels = TensorDataset(ind.apply_batch(labels_int.data).float(),
                           labels_int.runtime)
    return (feats
            .and_then(BlockLeastSquaresEstimator(conf.block_size,
                                                 conf.num_iter, conf.lam),
                      train, labels)
            .and_then(MaxClassifier()))


def build_mnist_random_cosine(train, labels_int,
                              conf: MnistConfig = MnistConfig(),
                              num_features: int = 4096,
                              gamma: float = 2e-3) -> Pipeline:
    """Driver config 2: MNIST RandomCosine features + BlockLeastSquares."""
    ind = ClassLabelIndicatorsFromIntLabels(conf.num_classes)
    labels = TensorDataset(ind.apply_batch(labels_int.data).float(),
                           labels_int.runtime)
    return (CosineRandomFeatures(conf.input_dim, num_features, gamma,
                                 seed=conf.seed)
            .and_then(BlockLeastSquaresEstimator(conf.block_size,
                                                 conf.num_iter, conf.lam),
                      train, labels)
            .and_then(MaxClassifier()))


@dataclass
class TimitConfig:
    input_dim: int = 440
    num_features: int = 8192     # reference headline runs used up to 200k
    gamma: float = 0.0555
    num_classes: int = 147
    block_size: int = 4096
    num_iter: int = 3
    lam: float = 1e-3
    seed: int = 0


def build_timit(train, labels_int, conf: TimitConfig = TimitConfig()
                ) -> Pipeline:
    ind = ClassLabelIndicatorsFromIntLabels(conf.num_classes)
    labels = TensorDataset(ind.apply_batch(labels_int.data).float(),
                           labels_int.runtime)
    return (StandardScaler().with_data(train)
            .and_then(CosineRandomFeatures(conf.input_dim, conf.num_features,
                                           conf.gamma, seed=conf.seed))
            .and_then(BlockLeastSquaresEstimator(conf.block_size,
                                                 conf.num_iter, conf.lam),
                      train, labels)
            .and_then(MaxClassifier()))
