from .sampler import DistributedSampler
from .synthetic import SyntheticImageDataset, make_dataset
from .loader import Loader

__all__ = ["DistributedSampler", "SyntheticImageDataset", "make_dataset", "Loader"]
