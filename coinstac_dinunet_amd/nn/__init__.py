"""Training core subpackage: NNTrainer epoch/eval loops (MI355X device
placement, fused-optimizer default)."""
from .basetrainer import NNTrainer

__all__ = ['NNTrainer']
