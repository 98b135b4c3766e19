from .basetrainer import NNTrainer

__all__ = ['NNTrainer']
