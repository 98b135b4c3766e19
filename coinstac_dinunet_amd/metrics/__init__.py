from .loss import dice_loss_binary
from .metrics import (AUCROCMetrics, COINNAverages, COINNMetrics,
                      ConfusionMatrix, Prf1a)

__all__ = ['COINNMetrics', 'COINNAverages', 'Prf1a', 'ConfusionMatrix',
           'AUCROCMetrics', 'dice_loss_binary']
