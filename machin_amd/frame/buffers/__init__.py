from .buffer import Buffer
from .prioritized_buffer import PrioritizedBuffer, WeightTree
from .storage import TransitionStorageBase, TransitionStorageBasic

__all__ = [
    "Buffer",
    "PrioritizedBuffer",
    "WeightTree",
    "TransitionStorageBase",
    "TransitionStorageBasic",
]
