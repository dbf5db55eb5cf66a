from .cosine_features import CosineRandomFeatures
from .fft import PaddedFFT
from .scaler import StandardScaler, StandardScalerModel
from .misc import (ColumnSampler, LinearRectifier, NormalizeRows,
                   RandomSignNode, Sampler, SignedHellingerMapper)

__all__ = [
    "CosineRandomFeatures", "PaddedFFT", "StandardScaler",
    "StandardScalerModel", "RandomSignNode", "LinearRectifier",
    "SignedHellingerMapper", "NormalizeRows", "ColumnSampler", "Sampler",
]
