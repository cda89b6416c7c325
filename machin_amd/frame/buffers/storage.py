"""Transition storage backends.

Parity target: reference ``machin/frame/buffers/storage.py``
(TransitionStorageBasic at :66) — a ring list of transitions moved to a
storage device.

MI355X note: the basic storage keeps transitions on the replay device
(CPU pinned or HBM). Unlike the reference it does NOT deepcopy python
objects — it clones tensors once during the device move, which is the
only copy the data path needs.
"""
from typing import Any, List, Union

import torch as t

from ..transition import TransitionBase


class TransitionStorageBase:
    """Interface: store transitions, return positions."""

    def store(self, transition: TransitionBase) -> Any:
        raise NotImplementedError

    def clear(self):
        raise NotImplementedError

    def __len__(self):
        raise NotImplementedError

    def __getitem__(self, item):
        raise NotImplementedError


class TransitionStorageBasic(TransitionStorageBase):
    """Ring storage of transition objects.

    Args:
        max_size: ring capacity.
        device: device the stored tensors live on ("cpu" keeps actor
            rollouts in host DRAM; a "cuda:N" device keeps the whole
            replay resident in HBM3E — 288 GB fits tens of millions of
            Atari frames, so prefer device residency on MI355X).
        pin_memory: pin CPU-stored tensors so the later H2D batch copy
            can run async on a side stream.
    """

    def __init__(
        self,
        max_size: int = 1000000,
        device: Union[str, t.device] = "cpu",
        pin_memory: bool = False,
    ):
        self.max_size = int(max_size)
        self.device = t.device(device)
        self.pin_memory = bool(pin_memory) and self.device.type == "cpu"
        self.data: List[TransitionBase] = []
        self.index = 0

    def store(self, transition: TransitionBase) -> int:
        transition = transition.clone()._detach().to(self.device)
        if self.pin_memory:
            for ma in transition.major_attr:
                d = getattr(transition, ma)
                for k, v in d.items():
                    d[k] = v.pin_memory()
        if len(self.data) < self.max_size:
            self.data.append(transition)
            return len(self.data) - 1
        pos = self.index
        self.data[pos] = transition
        self.index = (pos + 1) % self.max_size
        return pos

    def clear(self):
        self.data.clear()
        self.index = 0

    def __len__(self):
        return len(self.data)

    def __getitem__(self, item):
        return self.data[item]

    def __setitem__(self, key, value):
        self.data[key] = value
