from .data import (COINNDataHandle, COINNDataset, COINNPaddedDataSampler,
                   safe_collate)
from .datautils import (create_k_fold_splits, create_ratio_split, init_k_folds,
                        split_place_holder)

__all__ = ['COINNDataset', 'COINNDataHandle', 'COINNPaddedDataSampler',
           'safe_collate', 'create_k_fold_splits', 'create_ratio_split',
           'init_k_folds', 'split_place_holder']
