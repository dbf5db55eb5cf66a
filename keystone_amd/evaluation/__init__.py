from .classification import (BinaryClassifierEvaluator,
                             BinaryClassificationMetrics,
                             MulticlassClassifierEvaluator,
                             MulticlassMetrics)
from .mean_ap import MeanAveragePrecisionEvaluator
from .augmented import AugmentedExamplesEvaluator

__all__ = [
    "MulticlassClassifierEvaluator", "MulticlassMetrics",
    "BinaryClassifierEvaluator", "BinaryClassificationMetrics",
    "MeanAveragePrecisionEvaluator", "AugmentedExamplesEvaluator",
]
