from .data_generator import BaseGenerator, DataloaderGenerator, RandomTensorGenerator, RandomTokenGenerator
from .datasets import CIFAR10Dataset, GlueDataset, RandomMlpDataset, SyntheticGlueDataset

__all__ = [
    "BaseGenerator", "RandomTensorGenerator", "RandomTokenGenerator",
    "DataloaderGenerator", "RandomMlpDataset", "CIFAR10Dataset",
    "GlueDataset", "SyntheticGlueDataset",
]
