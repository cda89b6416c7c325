import torch as _t

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
    "DeviceTransitionBuffer",
    "default_buffer",
]


def default_buffer(
    replay_size: int,
    replay_device,
    prioritized: bool = False,
    **per_kwargs,
):
    """Pick the replay implementation for an algorithm's
    ``replay_device``: the HBM-resident flat rings (+ DeviceSumTree
    PER) when the replay lives on a GPU — the MI355X fast path is THE
    path — and the reference-style episodic Buffer / numpy-tree
    PrioritizedBuffer on CPU."""
    if _t.device(replay_device).type == "cuda":
        from .device_buffer import DeviceTransitionBuffer

        return DeviceTransitionBuffer(
            replay_size, replay_device, prioritized=prioritized,
            **per_kwargs,
        )
    if prioritized:
        return PrioritizedBuffer(replay_size, replay_device, **per_kwargs)
    return Buffer(replay_size, replay_device)


def __getattr__(name):
    # distributed buffers import the world lazily to keep plain buffer
    # use free of torch.distributed
    if name == "DistributedBuffer":
        from .buffer_d import DistributedBuffer

        return DistributedBuffer
    if name == "DistributedPrioritizedBuffer":
        from .prioritized_buffer_d import DistributedPrioritizedBuffer

        return DistributedPrioritizedBuffer
    if name == "DeviceTransitionBuffer":
        from .device_buffer import DeviceTransitionBuffer

        return DeviceTransitionBuffer
    raise AttributeError(name)
