from .synthetic import (synthetic_dense, synthetic_documents, synthetic_images,
                        synthetic_mnist, synthetic_timit)
from .files import (AmazonReviewsDataLoader, CifarLoader, CsvDataLoader,
                    ImageExtractor, ImageNetLoader, LabeledData,
                    LabelExtractor, NewsgroupsDataLoader,
                    TimitFeaturesDataLoader, VOCLoader)

__all__ = [
    "synthetic_dense", "synthetic_documents", "synthetic_images",
    "synthetic_mnist", "synthetic_timit",
    "CsvDataLoader", "CifarLoader", "NewsgroupsDataLoader",
    "AmazonReviewsDataLoader", "TimitFeaturesDataLoader", "ImageNetLoader",
    "VOCLoader", "LabeledData", "ImageExtractor", "LabelExtractor",
]
