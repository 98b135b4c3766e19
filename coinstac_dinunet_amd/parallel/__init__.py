from .cluster import (RcclCluster, load_cluster_state, save_cluster_state)
from .engine import (DEFAULT_BUCKET_BYTES, FlatGradBuffer, RcclLearner,
                     RcclReducer, init_distributed)
from .powersgd import RcclPowerSGDLearner, RcclPowerSGDReducer
from .rankdad import RcclDADLearner, RcclDADReducer

__all__ = ['RcclCluster', 'save_cluster_state', 'load_cluster_state',
           'FlatGradBuffer', 'RcclLearner', 'RcclReducer',
           'RcclPowerSGDLearner', 'RcclPowerSGDReducer',
           'RcclDADLearner', 'RcclDADReducer',
           'init_distributed', 'DEFAULT_BUCKET_BYTES']
