from .basic import (GrayScaler, ImageVectorizer, PixelScaler, Pooler,
                    SymmetricRectifier, FastWindower, Windower)
from .convolver import Convolver
from .patches import CenterCornerPatcher, RandomImageTransformer, RandomPatcher
from .sift import SIFTExtractor
from .lcs import LCSExtractor
from .daisy import DaisyExtractor
from .fisher_vector import FisherVector

__all__ = [
    "GrayScaler", "PixelScaler", "ImageVectorizer", "Windower", "Pooler",
    "SymmetricRectifier", "FastWindower", "Convolver", "CenterCornerPatcher",
    "RandomPatcher", "RandomImageTransformer", "SIFTExtractor",
    "LCSExtractor", "DaisyExtractor", "FisherVector",
]
