from .sampler import PKBatchSampler
from .synthetic import SyntheticImageDataset, SyntheticEmbeddingDataset
from .transforms import DataTransformer, TransformConfig
from .folder import FolderListDataset

__all__ = [
    "PKBatchSampler",
    "SyntheticImageDataset",
    "SyntheticEmbeddingDataset",
    "DataTransformer",
    "TransformConfig",
    "FolderListDataset",
]
