from .learner import COINNLearner
from .reducer import COINNReducer
from .nodes.local import COINNLocal
from .nodes.remote import COINNRemote

__all__ = ['COINNLearner', 'COINNReducer', 'COINNLocal', 'COINNRemote']
