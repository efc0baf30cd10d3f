from .sgd import FlatSGD
from .adam import FlatAdam

__all__ = ['FlatSGD', 'FlatAdam']
