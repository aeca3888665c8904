from .pipeline import load_and_preprocess_data
from .sampler import DistributedSampler
from .synthetic import SyntheticTextDataset, default_collate

__all__ = ["load_and_preprocess_data", "DistributedSampler",
           "SyntheticTextDataset", "default_collate"]
