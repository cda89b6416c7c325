"""Fixed-length sequence (RNN / R2D2-style) replay buffers.

Parity target: reference ``machin/frame/buffers/rnn_buffers.py``
(:19,190,259,415): sample fixed ``sample_length`` windows from stored
episodes; ``sample_dimension`` selects [batch, length, ...] (1) or
[length, batch, ...] (0) output layout via the
``post_process_attribute`` hook (:170-187); the prioritized variant
zeroes the priorities of episode-tail steps that cannot start a full
window (:364-380); distributed variants compose with the distributed
buffers.
"""
import random
from typing import List, Tuple, Union

import numpy as np
import torch as t

from .buffer import Buffer
from .buffer_d import DistributedBuffer
from .prioritized_buffer import PrioritizedBuffer
from .prioritized_buffer_d import DistributedPrioritizedBuffer


class _RNNWindowMixin:
    """Window expansion + reshaping shared by the RNN buffers."""

    def _init_rnn(self, sample_length: int, sample_dimension: int):
        if sample_length < 1:
            raise ValueError("sample_length must be >= 1.")
        if sample_dimension not in (0, 1):
            raise ValueError("sample_dimension must be 0 or 1.")
        self.sample_length = sample_length
        self.sample_dimension = sample_dimension
        self._last_window_num = 0

    # -- window helpers ------------------------------------------------
    def _valid_starts(self) -> List[Tuple[int, int]]:
        """All (episode_id, offset) pairs that can start a full
        window."""
        out = []
        for ep, handles in self.episode_transition_handles.items():
            for off in range(0, len(handles) - self.sample_length + 1):
                out.append((ep, off))
        return out

    def _window(self, episode_id: int, offset: int):
        handles = self.episode_transition_handles[episode_id]
        return [
            self.storage[h]
            for h in handles[offset : offset + self.sample_length]
        ]

    def _expand_starts(self, starts: List[Tuple[int, int]]):
        batch = []
        for ep, off in starts:
            batch.extend(self._window(ep, off))
        self._last_window_num = len(starts)
        return len(batch), batch

    # -- sampling ------------------------------------------------------
    def sample_method_random_unique(self, batch_size: int):
        starts = self._valid_starts()
        n = min(batch_size, len(starts))
        if n == 0:
            return 0, []
        return self._expand_starts(random.sample(starts, k=n))

    def sample_method_random(self, batch_size: int):
        starts = self._valid_starts()
        if not starts:
            return 0, []
        return self._expand_starts(
            [random.choice(starts) for _ in range(batch_size)]
        )

    def sample_method_all(self, _):
        return self._expand_starts(self._valid_starts())

    # -- reshaping -----------------------------------------------------
    def post_process_attribute(self, attr, key, tensor):
        if t.is_tensor(tensor) and self._last_window_num > 0:
            n, length = self._last_window_num, self.sample_length
            if tensor.shape[0] == n * length:
                tensor = tensor.view(n, length, *tensor.shape[1:])
                if self.sample_dimension == 0:
                    tensor = tensor.transpose(0, 1).contiguous()
        return tensor


class RNNBuffer(_RNNWindowMixin, Buffer):
    def __init__(
        self,
        sample_length: int,
        buffer_size: int = 1000000,
        buffer_device: Union[str, t.device] = "cpu",
        sample_dimension: int = 1,
        **kwargs,
    ):
        Buffer.__init__(self, buffer_size, buffer_device, **kwargs)
        self._init_rnn(sample_length, sample_dimension)


class RNNPrioritizedBuffer(_RNNWindowMixin, PrioritizedBuffer):
    """PER over window STARTS: tail steps that cannot begin a full
    window get (near-)zero priority so the tree never selects them."""

    def __init__(
        self,
        sample_length: int,
        buffer_size: int = 1000000,
        buffer_device: Union[str, t.device] = "cpu",
        sample_dimension: int = 1,
        **kwargs,
    ):
        PrioritizedBuffer.__init__(self, buffer_size, buffer_device, **kwargs)
        self._init_rnn(sample_length, sample_dimension)

    def store_episode(self, episode, priorities=None, required_attrs=(
            "state", "action", "next_state", "reward", "terminal")):
        handles = super().store_episode(
            episode, priorities=priorities, required_attrs=required_attrs
        )
        # zero the tail that cannot start a window
        tail = handles[max(len(handles) - self.sample_length + 1, 0):]
        if tail:
            self.wt_tree.update_leaf_batch(
                np.full(len(tail), 1e-12), np.asarray(tail, dtype=np.int64)
            )
        # remember handle -> (episode, offset) for window expansion
        return handles

    def sample_batch(self, batch_size: int, concatenate=True, device="cpu",
                     sample_attrs=None, additional_concat_custom_attrs=None,
                     *_, **__):
        if len(self.storage) == 0 or batch_size <= 0:
            return 0, None, None, None
        index, is_weight = self.sample_index_and_weight(batch_size)
        # expand each start handle into its window
        batch = []
        kept_index = []
        kept_weight = []
        for i, w in zip(index, is_weight):
            ep = self.transition_episode_number.get(int(i))
            if ep is None:
                continue
            handles = self.episode_transition_handles[ep]
            off = handles.index(int(i))
            if off + self.sample_length > len(handles):
                off = max(len(handles) - self.sample_length, 0)
            batch.extend(self._window(ep, off))
            kept_index.append(int(i))
            kept_weight.append(w)
        if not batch:
            return 0, None, None, None
        self._last_window_num = len(kept_index)
        result = self.post_process_batch(
            batch, device, concatenate, sample_attrs,
            additional_concat_custom_attrs,
        )
        return (
            len(kept_index),
            result,
            np.asarray(kept_index),
            np.asarray(kept_weight),
        )


class RNNDistributedBuffer(_RNNWindowMixin, DistributedBuffer):
    """Window sampling executed inside each member's local sample
    service (windows never split across members)."""

    def __init__(
        self,
        sample_length: int,
        buffer_name: str,
        group,
        buffer_size: int = 1000000,
        sample_dimension: int = 1,
        **kwargs,
    ):
        DistributedBuffer.__init__(self, buffer_name, group, buffer_size,
                                   **kwargs)
        self._init_rnn(sample_length, sample_dimension)

    def post_process_batch(self, batch, device, concatenate, sample_attrs,
                           additional_concat_custom_attrs):
        # batch arrives as a flat union of windows from members
        self._last_window_num = len(batch) // self.sample_length
        return super().post_process_batch(
            batch, device, concatenate, sample_attrs,
            additional_concat_custom_attrs,
        )


class RNNDistributedPrioritizedBuffer(
    _RNNWindowMixin, DistributedPrioritizedBuffer
):
    def __init__(
        self,
        sample_length: int,
        buffer_name: str,
        group,
        buffer_size: int = 1000000,
        sample_dimension: int = 1,
        **kwargs,
    ):
        DistributedPrioritizedBuffer.__init__(
            self, buffer_name, group, buffer_size, **kwargs
        )
        self._init_rnn(sample_length, sample_dimension)

    def store_episode(self, episode, priorities=None, required_attrs=(
            "state", "action", "next_state", "reward", "terminal")):
        handles = super().store_episode(
            episode, priorities=priorities, required_attrs=required_attrs
        )
        tail = handles[max(len(handles) - self.sample_length + 1, 0):]
        if tail:
            self.wt_tree.update_leaf_batch(
                np.full(len(tail), 1e-12), np.asarray(tail, dtype=np.int64)
            )
        return handles

    def _sample_service(self, batch_size: int, all_weight_sum: float,
                        sample_attrs=None,
                        additional_concat_custom_attrs=None):
        with self.wr_lock:
            if len(self.storage) == 0 or batch_size <= 0:
                return 0, None, None, None, None
            index, is_weight = self.sample_index_and_weight(
                batch_size, all_weight_sum, normalize=False
            )
            batch = []
            kept_index = []
            kept_weight = []
            for i, w in zip(index, is_weight):
                ep = self.transition_episode_number.get(int(i))
                if ep is None:
                    continue
                handles = self.episode_transition_handles[ep]
                off = handles.index(int(i))
                if off + self.sample_length > len(handles):
                    off = max(len(handles) - self.sample_length, 0)
                batch.extend(self._window(ep, off))
                kept_index.append(int(i))
                kept_weight.append(w)
            if not batch:
                return 0, None, None, None, None
            versions = self._entry_versions[np.asarray(kept_index,
                                                       dtype=np.int64)]
            result = self.post_process_batch(
                batch, "cpu", True, sample_attrs,
                additional_concat_custom_attrs,
            )
            return (
                len(kept_index),
                result,
                np.asarray(kept_index),
                np.asarray(kept_weight),
                versions,
            )

    def post_process_batch(self, batch, device, concatenate, sample_attrs,
                           additional_concat_custom_attrs):
        self._last_window_num = len(batch) // self.sample_length
        return super().post_process_batch(
            batch, device, concatenate, sample_attrs,
            additional_concat_custom_attrs,
        )
