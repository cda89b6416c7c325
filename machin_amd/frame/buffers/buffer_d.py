"""Distributed replay buffer.

Parity target: reference ``machin/frame/buffers/buffer_d.py`` (:17-198):
each group member holds a local buffer and registers size/clear/sample
services; ``sample_batch`` fans out async sample requests to every
member and concatenates the union. Storage is CPU-side (host DRAM is
the actor staging area on MI355X; learners move the concatenated batch
to HBM with one async H2D copy per attribute).
"""
import threading
from typing import Any, List, Union

import torch as t

from ...parallel.distributed.world import RpcGroup
from ..transition import TransitionBase
from .buffer import Buffer


def merge_processed_batches(parts: List[tuple], device) -> tuple:
    """Merge per-member concatenated batch tuples element-wise:
    tensors are concatenated along dim 0, per-key for tensor dicts;
    lists extend; wildcard dict-of-lists merge per key."""
    out = []
    for elems in zip(*parts):
        first = elems[0]
        if t.is_tensor(first):
            out.append(t.cat(list(elems), dim=0).to(device))
        elif isinstance(first, dict):
            if first and t.is_tensor(next(iter(first.values()))):
                out.append(
                    {
                        k: t.cat([e[k] for e in elems], dim=0).to(device)
                        for k in first.keys()
                    }
                )
            else:
                merged = {}
                for k in first.keys():
                    merged[k] = []
                    for e in elems:
                        merged[k].extend(e[k])
                out.append(merged)
        elif isinstance(first, list):
            merged_list = []
            for e in elems:
                merged_list.extend(e)
            out.append(merged_list)
        else:
            out.append(first)
    return tuple(out)


class DistributedBuffer(Buffer):
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
        me = group.get_cur_name()
        group.register(f"{buffer_name}/{me}/_size_service",
                       self._size_service)
        group.register(f"{buffer_name}/{me}/_clear_service",
                       self._clear_service)
        group.register(f"{buffer_name}/{me}/_sample_service",
                       self._sample_service)

    # -- local services ------------------------------------------------
    def _size_service(self) -> int:
        with self.wr_lock:
            return self.size()

    def _clear_service(self):
        with self.wr_lock:
            super().clear()
        return True

    def _sample_service(self, batch_size: int, sample_method: str,
                        sample_attrs=None,
                        additional_concat_custom_attrs=None):
        """Sample AND concatenate member-side: the reply is a handful
        of flat tensors per attribute instead of hundreds of python
        transition objects (an order of magnitude less
        (de)serialization on the learner's critical path)."""
        with self.wr_lock:
            if callable(sample_method):
                bsize, batch = sample_method(self, batch_size)
            else:
                method = getattr(self, "sample_method_" + sample_method)
                bsize, batch = method(batch_size)
            if bsize == 0 or not batch:
                return 0, None
            result = self.post_process_batch(
                batch, "cpu", True, sample_attrs,
                additional_concat_custom_attrs,
            )
            return bsize, result

    # -- writes (local) ------------------------------------------------
    def store_episode(self, episode, required_attrs=("state", "action",
                                                     "next_state", "reward",
                                                     "terminal")):
        with self.wr_lock:
            return super().store_episode(episode,
                                         required_attrs=required_attrs)

    def clear(self):
        """Clear only the LOCAL buffer."""
        with self.wr_lock:
            super().clear()

    def all_clear(self):
        """Clear every member's buffer."""
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
        sample_method: Union[str, Any] = "random_unique",
        sample_attrs: List[str] = None,
        additional_concat_custom_attrs: List[str] = None,
        *_,
        **__,
    ):
        members = self.group.get_group_members()
        # proportional allocation: ask everyone for an equal share
        per_member = max(1, batch_size // max(len(members), 1))
        futures = [
            self.group.registered_async(
                f"{self.buffer_name}/{m}/_sample_service",
                args=(per_member, sample_method, sample_attrs,
                      additional_concat_custom_attrs),
            )
            for m in members
        ]
        parts = []
        total = 0
        for f in futures:
            bsize, result = f.wait()
            if bsize > 0:
                parts.append(result)
                total += bsize
        if not parts:
            return 0, None
        return total, merge_processed_batches(parts, device)
