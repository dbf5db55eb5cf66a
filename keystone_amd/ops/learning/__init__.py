from .linear import LinearMapper, LinearMapEstimator, LocalLeastSquaresEstimator
from .block_ls import (BlockLinearMapper, BlockLeastSquaresEstimator,
                       BlockWeightedLeastSquaresEstimator, LazyBlocks)
from .least_squares import LeastSquaresEstimator
from .lbfgs import (DenseLBFGSwithL2, SparseLBFGSwithL2, MiniBatchSGDwithL2,
                    LeastSquaresGradient, LogisticGradient)
from .logistic import (LogisticRegressionEstimator, LogisticRegressionModel,
                       NaiveBayesEstimator, NaiveBayesModel)
from .kmeans import KMeansPlusPlusEstimator, KMeansModel
from .gmm import GaussianMixtureModel, GaussianMixtureModelEstimator
from .pca import (PCAEstimator, PCATransformer, DistributedPCAEstimator,
                  ApproximatePCAEstimator, TSQR)
from .whitening import ZCAWhitener, ZCAWhitenerEstimator, LinearDiscriminantAnalysis
from .krr import (GaussianKernelGenerator, KernelRidgeRegression,
                  KernelBlockLinearMapper)
from .kernel_matrix import KernelMatrix, BlockKernelMatrix

__all__ = [
    "LinearMapper", "LinearMapEstimator", "LocalLeastSquaresEstimator",
    "BlockLinearMapper", "BlockLeastSquaresEstimator",
    "BlockWeightedLeastSquaresEstimator", "LazyBlocks", "LeastSquaresEstimator",
    "DenseLBFGSwithL2", "SparseLBFGSwithL2", "MiniBatchSGDwithL2",
    "LeastSquaresGradient", "LogisticGradient",
    "LogisticRegressionEstimator", "LogisticRegressionModel",
    "NaiveBayesEstimator", "NaiveBayesModel",
    "KMeansPlusPlusEstimator", "KMeansModel",
    "GaussianMixtureModel", "GaussianMixtureModelEstimator",
    "PCAEstimator", "PCATransformer", "DistributedPCAEstimator",
    "ApproximatePCAEstimator", "TSQR",
    "ZCAWhitener", "ZCAWhitenerEstimator", "LinearDiscriminantAnalysis",
    "GaussianKernelGenerator", "KernelRidgeRegression",
    "KernelBlockLinearMapper", "KernelMatrix", "BlockKernelMatrix",
]
