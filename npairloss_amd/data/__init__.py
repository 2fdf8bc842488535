from .sampler import PKBatchSampler
from .synthetic import SyntheticImageDataset, SyntheticEmbeddingDataset
from .transforms import DataTransformer, TransformConfig

__all__ = [
    "PKBatchSampler",
    "SyntheticImageDataset",
    "SyntheticEmbeddingDataset",
    "DataTransformer",
    "TransformConfig",
]
