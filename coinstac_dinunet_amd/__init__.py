"""coinstac_dinunet_amd — MI355X-native decentralized federated training.

A from-scratch MI355X (gfx950/CDNA4) framework with the capabilities of
trendscenter/coinstac-dinunet v2.5.3: COINNLocal/COINNRemote phase machine,
COINNTrainer training core, k-fold/ratio splits, checkpoint-best selection,
site-weighted metric reduction, and three aggregation engines (dSGD,
PowerSGD, rankDAD) — with the file/JSON hand-off replaced by RCCL
collectives over xGMI (one process per GPU-site) and the hot ops running as
hand-written HIP kernels (coinstac_dinunet_amd.ops).
"""
import torch as _torch  # hard requirement, like the reference (__init__.py:1-9)

from .data import COINNDataset, COINNDataHandle
from .distrib import COINNLearner, COINNReducer, COINNLocal, COINNRemote
from .trainer import COINNTrainer

__version__ = '0.1.0'
__all__ = ['COINNDataset', 'COINNDataHandle', 'COINNLearner', 'COINNReducer',
           'COINNLocal', 'COINNRemote', 'COINNTrainer']
