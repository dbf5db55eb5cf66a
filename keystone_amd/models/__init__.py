from .text import (AmazonConfig, NewsgroupsConfig, build_amazon,
                   build_newsgroups)
from .dense import (MnistConfig, TimitConfig, build_mnist_randomfft,
                    build_mnist_random_cosine, build_timit)
from .imagenet_fv import ImageNetFVConfig, fit_imagenet_fv
from .cifar import CifarConfig, fit_cifar_patches
from .voc import VOCConfig, fit_voc

__all__ = [
    "NewsgroupsConfig", "build_newsgroups", "AmazonConfig", "build_amazon",
    "MnistConfig", "build_mnist_randomfft", "build_mnist_random_cosine", "TimitConfig", "build_timit",
    "ImageNetFVConfig", "fit_imagenet_fv", "CifarConfig", "fit_cifar_patches",
    "VOCConfig", "fit_voc",
]
