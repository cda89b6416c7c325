"""Prioritized experience replay: weight sum-tree + PER buffer.

Parity target: reference ``machin/frame/buffers/prioritized_buffer.py``
(WeightTree at :8, PrioritizedBuffer at :234): same public API
(``get_weight_sum / get_leaf_max / get_leaf_all_weights / get_leaf_weight /
find_leaf_index / update_leaf / update_leaf_batch / update_all_leaves``),
same PER semantics (α-normalized priorities, stratified segment
sampling, β-annealed importance weights, abs-TD priority update).

Design differences from the reference (not a port):

* heap layout — internal nodes occupy ``[1, capacity)`` and leaves
  ``[capacity, 2·capacity)`` in ONE flat float64 array; every level
  update is a single vectorized numpy pass, and the identical layout is
  what the gfx950 HIP sum-tree kernels (machin_amd.ops.sumtree) operate
  on, so CPU and GPU trees are drop-in interchangeable.
* batched find walks all queries down the levels simultaneously
  (vectorized compare/step), no python-per-query loop.

Reference CPU baselines to beat: build 10M leaves 90 ms,
10M lookups 230 ms, 1M batched update 20 ms on i7-6700HQ.
"""
import random
from typing import Any, List, Union

import numpy as np
import torch as t

from .buffer import Buffer
from .storage import TransitionStorageBase


class WeightTree:
    """Complete binary sum-tree over ``size`` leaves (float64)."""

    def __init__(self, size: int):
        if size <= 0:
            raise ValueError("Tree size must be positive.")
        self.size = int(size)
        self.depth = max(1, int(np.ceil(np.log2(max(self.size, 2)))))
        self.capacity = 1 << self.depth          # padded leaf count
        # heap: index 0 unused padding for root at 1
        self.weights = np.zeros(2 * self.capacity, dtype=np.float64)
        self.max_leaf = 0.0

    # -- accessors -----------------------------------------------------
    def get_weight_sum(self) -> float:
        return float(self.weights[1])

    def get_leaf_max(self) -> float:
        return float(self.max_leaf)

    def get_leaf_all_weights(self) -> np.ndarray:
        """View of the ``size`` real leaf weights."""
        return self.weights[self.capacity : self.capacity + self.size]

    def get_leaf_weight(
        self, index: Union[int, List[int], np.ndarray]
    ) -> Any:
        index = np.asarray(index)
        if np.any(index >= self.size) or np.any(index < 0):
            raise ValueError("Index out of range.")
        out = self.weights[self.capacity + index]
        if out.ndim == 0:
            return float(out)
        return out

    # -- queries -------------------------------------------------------
    def find_leaf_index(
        self, weight: Union[float, List[float], np.ndarray]
    ) -> Union[int, np.ndarray]:
        """Map prefix-sum weights to leaf indexes (vectorized walk)."""
        scalar = np.isscalar(weight) or (
            isinstance(weight, np.ndarray) and weight.ndim == 0
        )
        w = np.atleast_1d(np.asarray(weight, dtype=np.float64)).copy()
        node = np.ones(len(w), dtype=np.int64)  # start at root
        for _ in range(self.depth):
            left = node << 1
            left_w = self.weights[left]
            go_right = w > left_w
            w = np.where(go_right, w - left_w, w)
            node = left + go_right
        index = np.minimum(node - self.capacity, self.size - 1)
        index = np.maximum(index, 0)
        if scalar:
            return int(index[0])
        return index

    # -- updates -------------------------------------------------------
    def update_leaf(self, weight: float, index: int):
        """Set one leaf and propagate to the root."""
        if not 0 <= index < self.size:
            raise ValueError("Index out of range.")
        weight = float(weight)
        self.max_leaf = max(self.max_leaf, weight)
        node = self.capacity + index
        delta = weight - self.weights[node]
        self.weights[node] = weight
        node >>= 1
        while node >= 1:
            self.weights[node] += delta
            node >>= 1

    def update_leaf_batch(
        self,
        weights: Union[List[float], np.ndarray],
        indexes: Union[List[int], np.ndarray],
    ):
        """Set many leaves and repair ancestors level by level."""
        weights = np.asarray(weights, dtype=np.float64)
        indexes = np.asarray(indexes, dtype=np.int64)
        if weights.shape != indexes.shape:
            raise ValueError("weights and indexes must have the same length.")
        if weights.size == 0:
            return
        if np.any(indexes >= self.size) or np.any(indexes < 0):
            raise ValueError("Index out of range.")
        if weights.size:
            self.max_leaf = max(self.max_leaf, float(weights.max()))
        # last write wins for duplicate indexes
        self.weights[self.capacity + indexes] = weights
        parents = np.unique((self.capacity + indexes) >> 1)
        while parents.size and parents[0] >= 1:
            left = parents << 1
            self.weights[parents] = self.weights[left] + self.weights[left + 1]
            parents = np.unique(parents >> 1)
            if parents[0] == 0:
                parents = parents[1:]

    def update_all_leaves(self, weights: Union[List[float], np.ndarray]):
        """Replace every leaf and rebuild the tree bottom-up."""
        weights = np.asarray(weights, dtype=np.float64)
        if weights.size != self.size:
            raise ValueError(
                f"Must provide {self.size} weights, got {weights.size}."
            )
        self.weights[self.capacity : self.capacity + self.size] = weights
        self.max_leaf = float(weights.max()) if weights.size else 0.0
        self._build()

    def _build(self):
        """Bottom-up rebuild: one vectorized add per level."""
        lo, n = self.capacity, self.capacity
        while n > 1:
            level = self.weights[lo : lo + n]
            n >>= 1
            lo >>= 1
            self.weights[lo : lo + n] = level[0::2] + level[1::2]

    def print_weights(self, precision: int = 2):
        fmt = f"{{:.{precision}f}}"
        lo, n = 1, 1
        while lo <= self.capacity:
            print(" ".join(fmt.format(v) for v in self.weights[lo : lo + n]))
            lo <<= 1
            n <<= 1


class PrioritizedBuffer(Buffer):
    """PER buffer: stratified sampling by priority with IS weights."""

    def __init__(
        self,
        buffer_size: int = 1000000,
        buffer_device: Union[str, t.device] = "cpu",
        epsilon: float = 1e-2,
        alpha: float = 0.6,
        beta: float = 0.4,
        beta_increment_per_sampling: float = 0.001,
        storage: TransitionStorageBase = None,
        **kwargs,
    ):
        super().__init__(
            buffer_size=buffer_size,
            buffer_device=buffer_device,
            storage=storage,
            **kwargs,
        )
        self.epsilon = epsilon
        self.alpha = alpha
        self.beta = beta
        self.beta_increment_per_sampling = beta_increment_per_sampling
        self.curr_beta = beta
        self.wt_tree = WeightTree(self.storage.max_size)

    def _normalize_priority(self, priority):
        return (np.abs(np.asarray(priority, dtype=np.float64)) + self.epsilon) ** (
            self.alpha
        )

    def store_episode(
        self,
        episode,
        priorities: Union[List[float], np.ndarray] = None,
        required_attrs=("state", "action", "next_state", "reward", "terminal"),
    ):
        handles = super().store_episode(episode, required_attrs=required_attrs)
        if priorities is None:
            # new samples get max priority so each is seen at least once
            max_leaf = self.wt_tree.get_leaf_max()
            prio = np.full(
                len(handles), max_leaf if max_leaf > 0 else 1.0, dtype=np.float64
            )
        else:
            prio = self._normalize_priority(priorities)
            if prio.size != len(handles):
                raise ValueError("One priority per transition required.")
        self.wt_tree.update_leaf_batch(prio, np.asarray(handles, dtype=np.int64))
        return handles

    def clear(self):
        super().clear()
        self.wt_tree = WeightTree(self.storage.max_size)
        self.curr_beta = self.beta

    def update_priority(self, priorities: np.ndarray, indexes: np.ndarray):
        """Write back new (abs-TD) priorities for sampled transitions."""
        self.wt_tree.update_leaf_batch(
            self._normalize_priority(priorities), np.asarray(indexes, dtype=np.int64)
        )

    def sample_index_and_weight(
        self, batch_size: int, all_weight_sum: float = None,
        normalize: bool = True,
    ):
        """Stratified segment sampling + β-annealed IS weights.
        ``normalize=False`` returns raw IS weights (the distributed
        buffer normalizes once across all members)."""
        segment_sum = self.wt_tree.get_weight_sum()
        if all_weight_sum is None:
            all_weight_sum = segment_sum
        seg_len = segment_sum / batch_size
        rand = (
            np.random.uniform(size=batch_size) + np.arange(batch_size, dtype=np.float64)
        ) * seg_len
        index = self.wt_tree.find_leaf_index(rand)
        # clamp to stored region (tree leaves beyond size are zero weight)
        index = np.minimum(index, max(len(self.storage) - 1, 0))
        leaf_weight = self.wt_tree.get_leaf_weight(index)
        # probability relative to the GLOBAL weight sum (matters for the
        # distributed buffer where segments span processes)
        probs = np.maximum(leaf_weight / all_weight_sum, 1e-12)
        is_weight = np.power(len(self.storage) * probs, -self.curr_beta)
        if normalize:
            is_weight /= is_weight.max()
        self.curr_beta = min(
            1.0, self.curr_beta + self.beta_increment_per_sampling
        )
        return index, is_weight

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
        """Returns ``(real_batch_size, batch, indexes, is_weights)``."""
        if len(self.storage) == 0 or batch_size <= 0:
            return 0, None, None, None
        index, is_weight = self.sample_index_and_weight(batch_size)
        batch = [self.storage[int(i)] for i in index]
        result = self.post_process_batch(
            batch,
            device=device,
            concatenate=concatenate,
            sample_attrs=sample_attrs,
            additional_concat_custom_attrs=additional_concat_custom_attrs,
        )
        return batch_size, result, index, is_weight
