from .datasets import (ArrayDataset, TensorDataset, H5Dataset, get_datasets,
                       NORM_STATS)
from .partition import DatasetSplit, distribute_data
from .poison import poison_dataset, pattern_spec, apply_pattern_, PatternSpec

__all__ = ['ArrayDataset', 'TensorDataset', 'H5Dataset', 'get_datasets',
           'NORM_STATS', 'DatasetSplit', 'distribute_data', 'poison_dataset',
           'pattern_spec', 'apply_pattern_', 'PatternSpec']
