from .augmentations import DataAugmentationDINO
from .collate import collate_data_and_cast, get_batch_subset
from .datasets import ADE20K, CocoCaptions, ImageNet, ImageNet22k, Split, SyntheticDataset
from .loaders import SamplerType, make_data_loader, make_dataset
from .masking import MaskingGenerator
from .samplers import EpochSampler, InfiniteSampler, ShardedInfiniteSampler

__all__ = [
    "DataAugmentationDINO",
    "collate_data_and_cast",
    "get_batch_subset",
    "MaskingGenerator",
    "make_dataset",
    "make_data_loader",
    "SamplerType",
    "EpochSampler",
    "InfiniteSampler",
    "ShardedInfiniteSampler",
    "ImageNet",
    "ImageNet22k",
    "ADE20K",
    "CocoCaptions",
    "SyntheticDataset",
    "Split",
]
