from .labels import (ClassLabelIndicatorsFromIntLabels,
                     ClassLabelIndicatorsFromStringLabels, MaxClassifier,
                     TopKClassifier)
from .vectors import (Cast, Densify, Sparsify, VectorCombiner, VectorSplitter)
from .misc import Shuffler, WeightedNode

__all__ = [
    "ClassLabelIndicatorsFromIntLabels", "ClassLabelIndicatorsFromStringLabels",
    "MaxClassifier", "TopKClassifier", "VectorCombiner", "VectorSplitter",
    "Densify", "Sparsify", "Cast", "Shuffler", "WeightedNode",
]
