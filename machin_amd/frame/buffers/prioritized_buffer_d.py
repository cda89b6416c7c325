"""Distributed prioritized replay.

Parity target: reference ``machin/frame/buffers/prioritized_buffer_d.py``
(:11-303): a local weight-tree PER buffer per member plus an
entry-version table that rejects stale priority updates (:282-291);
global sampling gathers per-member weight sums, allocates sample
counts proportionally (:220-245), merges samples and IS weights, and
routes priority updates back to the owning members (:184-202).
"""
import threading
from typing import List, Union

import numpy as np
import torch as t

from ...parallel.distributed.world import RpcGroup
from .prioritized_buffer import PrioritizedBuffer


class DistributedPrioritizedBuffer(PrioritizedBuffer):
    def __init__(
        self,
        buffer_name: str,
        group: RpcGroup,
        buffer_size: int = 1000000,
        *_,
        **kwargs,
    ):
        super().__init__(buffer_size=buffer_size, buffer_device="cpu",
                         **kwargs)
        self.buffer_name = buffer_name
        self.group = group
        self.wr_lock = threading.RLock()
        # version per storage slot: stale priority updates (for an
        # entry that has been overwritten since sampling) are dropped
        self._entry_versions = np.zeros(
            self.storage.max_size, dtype=np.int64
        )
        me = group.get_cur_name()
        group.register(f"{buffer_name}/{me}/_weight_sum_service",
                       self._weight_sum_service)
        group.register(f"{buffer_name}/{me}/_size_service",
                       self._size_service)
        group.register(f"{buffer_name}/{me}/_clear_service",
                       self._clear_service)
        group.register(f"{buffer_name}/{me}/_sample_service",
                       self._sample_service)
        group.register(f"{buffer_name}/{me}/_update_priority_service",
                       self._update_priority_service)

    # -- local services ------------------------------------------------
    def _weight_sum_service(self) -> float:
        with self.wr_lock:
            return self.wt_tree.get_weight_sum()

    def _size_service(self) -> int:
        with self.wr_lock:
            return len(self.storage)

    def _clear_service(self):
        with self.wr_lock:
            PrioritizedBuffer.clear(self)
            self._entry_versions[:] = 0
        return True

    def _sample_service(self, batch_size: int, all_weight_sum: float,
                        sample_attrs=None,
                        additional_concat_custom_attrs=None):
        """Member-side: sample by priority AND concatenate; replies
        carry flat per-attribute tensors (see buffer_d)."""
        with self.wr_lock:
            if len(self.storage) == 0 or batch_size <= 0:
                return 0, None, None, None, None
            index, is_weight = self.sample_index_and_weight(
                batch_size, all_weight_sum, normalize=False
            )
            batch = [self.storage[int(i)] for i in index]
            versions = self._entry_versions[index].copy()
            result = self.post_process_batch(
                batch, "cpu", True, sample_attrs,
                additional_concat_custom_attrs,
            )
            return batch_size, result, index, is_weight, versions

    def _update_priority_service(self, priorities, indexes, versions):
        with self.wr_lock:
            fresh = self._entry_versions[indexes] == versions
            if fresh.any():
                self.wt_tree.update_leaf_batch(
                    self._normalize_priority(priorities[fresh]),
                    np.asarray(indexes)[fresh],
                )
        return True

    # -- writes (local) ------------------------------------------------
    def store_episode(self, episode, priorities=None,
                      required_attrs=("state", "action", "next_state",
                                      "reward", "terminal")):
        with self.wr_lock:
            handles = super().store_episode(
                episode, priorities=priorities, required_attrs=required_attrs
            )
            self._entry_versions[np.asarray(handles)] += 1
            return handles

    def clear(self):
        with self.wr_lock:
            PrioritizedBuffer.clear(self)
            self._entry_versions[:] = 0

    def all_clear(self):
        futures = [
            self.group.registered_async(
                f"{self.buffer_name}/{m}/_clear_service"
            )
            for m in self.group.get_group_members()
        ]
        for f in futures:
            f.wait()

    def size(self) -> int:
        return len(self.storage)

    def all_size(self) -> int:
        futures = [
            self.group.registered_async(
                f"{self.buffer_name}/{m}/_size_service"
            )
            for m in self.group.get_group_members()
        ]
        return sum(f.wait() for f in futures)

    # -- global sampling -----------------------------------------------
    def sample_batch(
        self,
        batch_size: int,
        concatenate: bool = True,
        device: Union[str, t.device] = "cpu",
        sample_attrs: List[str] = None,
        additional_concat_custom_attrs: List[str] = None,
        *_,
        **__,
    ):
        members = self.group.get_group_members()
        sum_futures = [
            self.group.registered_async(
                f"{self.buffer_name}/{m}/_weight_sum_service"
            )
            for m in members
        ]
        weight_sums = [f.wait() for f in sum_futures]
        all_weight_sum = float(sum(weight_sums))
        if all_weight_sum <= 0 or batch_size <= 0:
            return 0, None, None, None
        # allocate per-member counts proportional to weight mass
        counts = [
            int(round(batch_size * ws / all_weight_sum))
            for ws in weight_sums
        ]
        # fix rounding drift
        while sum(counts) < batch_size:
            counts[int(np.argmax(weight_sums))] += 1
        sample_futures = [
            (
                m,
                self.group.registered_async(
                    f"{self.buffer_name}/{m}/_sample_service",
                    args=(c, all_weight_sum, sample_attrs,
                          additional_concat_custom_attrs),
                ),
            )
            for m, c in zip(members, counts)
            if c > 0
        ]
        parts = []
        all_index = {}
        all_is_weight = []
        total = 0
        for m, f in sample_futures:
            bsize, result, index, is_weight, versions = f.wait()
            if bsize > 0:
                parts.append(result)
                total += bsize
                all_index[m] = (index, versions)
                all_is_weight.append(is_weight)
        if not parts:
            return 0, None, None, None
        is_weight = np.concatenate(all_is_weight)
        # renormalize IS weights globally
        is_weight = is_weight / max(is_weight.max(), 1e-12)
        from .buffer_d import merge_processed_batches

        result = merge_processed_batches(parts, device)
        return total, result, all_index, is_weight

    def update_priority(self, priorities: np.ndarray, indexes: dict):
        """Route new priorities back to each owning member;
        ``indexes`` is the map returned by sample_batch."""
        priorities = np.asarray(priorities, dtype=np.float64)
        offset = 0
        futures = []
        for m, (index, versions) in indexes.items():
            n = len(index)
            futures.append(
                self.group.registered_async(
                    f"{self.buffer_name}/{m}/_update_priority_service",
                    args=(priorities[offset : offset + n], index, versions),
                )
            )
            offset += n
        for f in futures:
            f.wait()
