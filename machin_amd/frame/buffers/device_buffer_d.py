"""Distributed prioritized replay with learner-resident HBM shards —
the MI355X redesign of the reference's sample-side-pull buffer.

Reference (`machin/frame/buffers/prioritized_buffer_d.py:11`): every
process keeps a LOCAL CPU shard + weight tree; the learner's
``sample_batch`` fans RPC calls to all members and ships pickled
transition lists back on every update — the hot sampling path crosses
process boundaries.

Here the data flows the other way, once: sampler processes flatten
each episode to a dict of flat CPU tensors and push it (round-robin)
to ONE learner member over the control plane; the learner stages the
tensors through a reusable pinned slab and issues one async H2D copy
per attribute into its :class:`DeviceTransitionBuffer` HBM rings.
Sampling, IS weights and priority updates are then entirely local to
the learner's GPU (DeviceSumTree kernels) — zero RPC and zero host
round trips on the per-update path. With multiple learners each hosts
an independent shard fed a disjoint round-robin split of the episode
stream, so per-shard PER statistics remain uniform.
"""
from typing import Dict, List, Union

import torch as t

from ...parallel.distributed.world import RpcGroup
from .device_buffer import DeviceTransitionBuffer


class DeviceDistributedPrioritizedBuffer:
    """Store-side-push distributed PER over learner HBM shards."""

    accepts_tensor_priorities = True

    def __init__(
        self,
        buffer_name: str,
        group: RpcGroup,
        buffer_size: int,
        learners: List[str] = None,
        device: Union[str, t.device] = "cuda:0",
        epsilon: float = 1e-2,
        alpha: float = 0.6,
        beta: float = 0.4,
        beta_increment_per_sampling: float = 0.001,
    ):
        self.buffer_name = buffer_name
        self.group = group
        members = group.get_group_members()
        self.learners = list(learners) if learners else members[:1]
        for ln in self.learners:
            if ln not in members:
                raise ValueError(f"Learner {ln!r} not in group members.")
        me = group.get_cur_name()
        self.me = me
        self.is_learner = me in self.learners
        self.local: DeviceTransitionBuffer = None
        if self.is_learner:
            self.local = DeviceTransitionBuffer(
                buffer_size, device, prioritized=True,
                epsilon=epsilon, alpha=alpha, beta=beta,
                beta_increment_per_sampling=beta_increment_per_sampling,
            )
            group.register(
                f"{buffer_name}/{me}/_store_service", self._store_service
            )
            group.register(
                f"{buffer_name}/{me}/_size_service", self._size_service
            )
            group.register(
                f"{buffer_name}/{me}/_clear_service", self._clear_service
            )
        # flattening helper for non-learner members (never allocates
        # device memory)
        self._flattener = DeviceTransitionBuffer(1, "cpu")
        self._rr = 0

    # -- services (run on learner members) -----------------------------
    def _store_service(self, batch: Dict[str, t.Tensor]):
        self.local.store_flat(batch)
        return True

    def _size_service(self) -> int:
        return self.local.size()

    def _clear_service(self):
        self.local.clear()
        return True

    # -- storing (any member) ------------------------------------------
    def store_episode(
        self,
        episode,
        required_attrs=("state", "action", "next_state", "reward",
                        "terminal"),
        **__,
    ):
        batch = self._flattener.flatten_episode(episode, required_attrs)
        target = self.learners[self._rr % len(self.learners)]
        self._rr += 1
        if target == self.me:
            self.local.store_flat(batch)
        else:
            self.group.registered_sync(
                f"{self.buffer_name}/{target}/_store_service",
                args=(batch,),
            )

    def append(self, transition, *_, **__):
        self.store_episode([transition], required_attrs=())

    # -- sampling / priorities (learner-local) --------------------------
    def sample_batch(self, batch_size: int, concatenate: bool = True,
                     device=None, sample_attrs=None, *_, **__):
        if not self.is_learner:
            return 0, None, None, None
        return self.local.sample_batch(
            batch_size, concatenate, device, sample_attrs=sample_attrs
        )

    def update_priority(self, priorities, indexes):
        if not self.is_learner:
            raise RuntimeError(
                "update_priority is learner-local; this member hosts "
                "no shard."
            )
        self.local.update_priority(priorities, indexes)

    # -- bookkeeping ----------------------------------------------------
    def size(self) -> int:
        return self.local.size() if self.is_learner else 0

    def all_size(self) -> int:
        futures = [
            self.group.registered_async(
                f"{self.buffer_name}/{ln}/_size_service"
            )
            for ln in self.learners
        ]
        return sum(f.wait() for f in futures)

    def clear(self):
        for ln in self.learners:
            self.group.registered_sync(
                f"{self.buffer_name}/{ln}/_clear_service"
            )

    def __len__(self):
        return self.size()
