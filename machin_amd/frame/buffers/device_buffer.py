"""HBM-resident replay buffers (MI355X fast path).

No reference counterpart — the reference stores replay as python
Transition objects on CPU (machin/frame/buffers/storage.py) and pays
python/copy costs per sample. On MI355X, 288 GB HBM3E fits tens of
millions of Atari transitions, so the fast path keeps the whole
replay as preallocated flat device rings:

* ``store_batch``: one async H2D copy per attribute (from pinned host
  staging or device tensors);
* ``sample_batch``: device-side ``index_select`` per attribute — zero
  python-per-transition work, zero host round trips;
* ``DevicePrioritizedBuffer``: priorities in a
  machin_amd.ops.sumtree.DeviceSumTree — stratified sampling and
  priority updates never leave the GPU.
"""
from typing import Dict, Tuple, Union

import torch as t


class DeviceReplayBuffer:
    """Uniform-sampling flat ring buffer on a ROCm device."""

    def __init__(
        self,
        buffer_size: int,
        spec: Dict[str, Tuple[Tuple[int, ...], t.dtype]],
        device: Union[str, t.device] = "cuda:0",
    ):
        """``spec`` maps attribute name -> (per-item shape, dtype);
        e.g. {"state": ((4, 84, 84), torch.uint8), "reward": ((),
        torch.float32)}."""
        self.capacity = int(buffer_size)
        self.device = t.device(device)
        self.spec = dict(spec)
        self.data = {
            k: t.empty((self.capacity, *shape), dtype=dtype,
                       device=self.device)
            for k, (shape, dtype) in self.spec.items()
        }
        self._size = 0
        self._head = 0

    def size(self) -> int:
        return self._size

    def clear(self):
        self._size = 0
        self._head = 0

    def store_batch(self, batch: Dict[str, t.Tensor]) -> t.Tensor:
        """Store n items; returns their ring positions (device
        int64)."""
        keys = set(batch.keys())
        if keys != set(self.spec.keys()):
            raise ValueError(
                f"Batch keys {sorted(keys)} != spec keys "
                f"{sorted(self.spec)}"
            )
        n = next(iter(batch.values())).shape[0]
        if n > self.capacity:
            raise ValueError("Batch larger than buffer capacity.")
        end = self._head + n
        pos = (
            t.arange(self._head, end, device=self.device) % self.capacity
        )
        for k, v in batch.items():
            v = v.to(self.device, non_blocking=True)
            if end <= self.capacity:
                self.data[k][self._head : end] = v
            else:
                split = self.capacity - self._head
                self.data[k][self._head :] = v[:split]
                self.data[k][: end % self.capacity] = v[split:]
        self._head = end % self.capacity
        self._size = min(self._size + n, self.capacity)
        return pos

    def sample_batch(self, batch_size: int) -> Dict[str, t.Tensor]:
        """Uniform sample with replacement; everything on device."""
        if self._size == 0:
            return {}
        idx = t.randint(
            0, self._size, (batch_size,), device=self.device
        )
        return self.gather(idx)

    def gather(self, idx: t.Tensor) -> Dict[str, t.Tensor]:
        return {
            k: buf.index_select(0, idx) for k, buf in self.data.items()
        }


class DevicePrioritizedBuffer(DeviceReplayBuffer):
    """PER on device: DeviceSumTree priorities, stratified sampling
    and IS weights computed by gfx950 kernels."""

    def __init__(
        self,
        buffer_size: int,
        spec: Dict[str, Tuple[Tuple[int, ...], t.dtype]],
        device: Union[str, t.device] = "cuda:0",
        epsilon: float = 1e-2,
        alpha: float = 0.6,
        beta: float = 0.4,
        beta_increment_per_sampling: float = 0.001,
    ):
        super().__init__(buffer_size, spec, device)
        from ...ops.sumtree import DeviceSumTree

        self.wt_tree = DeviceSumTree(self.capacity, self.device)
        self.epsilon = epsilon
        self.alpha = alpha
        self.curr_beta = beta
        self.beta_increment_per_sampling = beta_increment_per_sampling
        self._max_priority = t.ones((), device=self.device)

    def _normalize(self, priorities: t.Tensor) -> t.Tensor:
        return (priorities.abs() + self.epsilon) ** self.alpha

    def store_batch(self, batch: Dict[str, t.Tensor],
                    priorities: t.Tensor = None) -> t.Tensor:
        pos = super().store_batch(batch)
        if priorities is None:
            prio = self._max_priority.expand(pos.numel()).contiguous()
        else:
            prio = self._normalize(
                priorities.to(self.device, non_blocking=True).float()
            )
            self._max_priority = t.maximum(
                self._max_priority, prio.max()
            )
        self.wt_tree.update_leaf_batch(prio, pos)
        return pos

    def sample_batch(self, batch_size: int):
        """Returns (batch dict, indexes, is_weights) — all on
        device, no host synchronization."""
        if self._size == 0:
            return {}, None, None
        idx = self.wt_tree.sample(batch_size, stratified=True)
        idx = idx.clamp_max(self._size - 1)
        leaf_w = self.wt_tree.get_leaf_weight(idx)
        total = self.wt_tree.get_weight_sum_tensor()
        probs = (leaf_w / total).clamp_min(1e-12)
        is_weight = (self._size * probs) ** (-self.curr_beta)
        is_weight = is_weight / is_weight.max()
        self.curr_beta = min(
            1.0, self.curr_beta + self.beta_increment_per_sampling
        )
        return self.gather(idx), idx, is_weight

    def update_priority(self, priorities: t.Tensor, idx: t.Tensor):
        prio = self._normalize(priorities.float())
        self._max_priority = t.maximum(self._max_priority, prio.max())
        self.wt_tree.update_leaf_batch(prio, idx)

    def clear(self):
        super().clear()
        self.wt_tree.weights.zero_()
        self._max_priority = t.ones((), device=self.device)
