from .buffer import Buffer
from .prioritized_buffer import PrioritizedBuffer, WeightTree
from .storage import TransitionStorageBase, TransitionStorageBasic

__all__ = [
    "Buffer",
    "PrioritizedBuffer",
    "WeightTree",
    "TransitionStorageBase",
    "TransitionStorageBasic",
    "DistributedBuffer",
    "DistributedPrioritizedBuffer",
]


def __getattr__(name):
    # distributed buffers import the world lazily to keep plain buffer
    # use free of torch.distributed
    if name == "DistributedBuffer":
        from .buffer_d import DistributedBuffer

        return DistributedBuffer
    if name == "DistributedPrioritizedBuffer":
        from .prioritized_buffer_d import DistributedPrioritizedBuffer

        return DistributedPrioritizedBuffer
    raise AttributeError(name)
