from .data import (COINNDataHandle, COINNDataset, COINNPaddedDataSampler,
                   safe_collate)
from .datautils import (create_k_fold_splits, create_ratio_split, init_k_folds,
                        split_place_holder)

__all__ = ['COINNDataset', 'COINNDataHandle', 'COINNPaddedDataSampler',
           'safe_collate', 'create_k_fold_splits', 'create_ratio_split',
           'init_k_folds', 'split_place_holder']

# the reference data namespace re-exports the enums and logger helpers
from ..config.keys import AGG_Engine, GatherMode, Key, Mode, Phase  # noqa: E402,F401
from ..utils.logger import error, info, success, warn  # noqa: E402,F401
from ..utils import lazy_debug  # noqa: E402,F401
