"""ARS: augmented random search (gradient-free, distributed).

Parity target: reference ``machin/frame/algorithms/ars.py``:
Welford ``RunningStat`` / ``MeanStdFilter`` state normalization with
local/global split (:24-242), shared noise table + seeded samplers
(:245-268), rollout assignment by worker index (:332-339), update =
pair rollout rewards to the manager, top-percentile direction
selection, reward-diff-weighted noise sum as the gradient
(:504-601), filter collect/sync via pair/barrier (:640-655), model
push/pull server for parameters.
"""
from typing import Any, Dict, List, Tuple

import numpy as np
import torch as t
import torch.nn as nn

from ...parallel.distributed.world import RpcGroup
from ...parallel.server.param_server import PushPullModelServer
from .base import TorchFramework
from .utils import safe_call, safe_return


class RunningStat:
    """Welford online mean/variance over tensors of a fixed shape."""

    def __init__(self, shape):
        self._n = 0
        self._mean = t.zeros(shape, dtype=t.float64)
        self._m2 = t.zeros(shape, dtype=t.float64)
        self._shape = tuple(shape) if not isinstance(shape, tuple) else shape

    def copy(self):
        out = RunningStat(self._shape)
        out._n = self._n
        out._mean = self._mean.clone()
        out._m2 = self._m2.clone()
        return out

    def push(self, x: t.Tensor):
        x = x.to(t.float64)
        if tuple(x.shape) != tuple(self._shape):
            raise ValueError(
                f"Shape mismatch: expected {self._shape}, got {tuple(x.shape)}"
            )
        self._n += 1
        delta = x - self._mean
        self._mean = self._mean + delta / self._n
        self._m2 = self._m2 + delta * (x - self._mean)

    def update(self, other: "RunningStat"):
        """Merge another RunningStat (parallel Welford combine)."""
        n1, n2 = self._n, other._n
        if n2 == 0:
            return
        if n1 == 0:
            self._n = other._n
            self._mean = other._mean.clone()
            self._m2 = other._m2.clone()
            return
        delta = other._mean - self._mean
        n = n1 + n2
        self._mean = self._mean + delta * (n2 / n)
        self._m2 = self._m2 + other._m2 + delta.square() * (n1 * n2 / n)
        self._n = n

    @property
    def n(self):
        return self._n

    @property
    def mean(self):
        return self._mean

    @property
    def var(self):
        if self._n <= 1:
            return self._mean.square()
        return self._m2 / (self._n - 1)

    @property
    def std(self):
        return self.var.sqrt()

    @property
    def shape(self):
        return self._shape

    def __repr__(self):
        return f"RunningStat(n={self._n}, shape={self._shape})"


class MeanStdFilter:
    """State normalizer with a local buffer merged into global stats
    at sync time (reference :132-242)."""

    def __init__(self, shape):
        self.shape = shape
        self.rs = RunningStat(shape)         # global (applied)
        self.buffer = RunningStat(shape)     # local since last sync
        self.mean = t.zeros(shape, dtype=t.float64)
        self.std = t.ones(shape, dtype=t.float64)

    def clear_local(self):
        self.buffer = RunningStat(self.shape)

    def copy(self):
        out = MeanStdFilter(self.shape)
        out.rs = self.rs.copy()
        out.buffer = self.buffer.copy()
        out.mean = self.mean.clone()
        out.std = self.std.clone()
        return out

    def collect(self, other: "MeanStdFilter"):
        """Merge another filter's LOCAL buffer into our global stats."""
        self.rs.update(other.buffer)

    def sync(self, other: "MeanStdFilter"):
        """Adopt another filter's global stats."""
        self.rs = other.rs.copy()

    def apply_stats(self):
        self.mean = self.rs.mean.clone()
        std = self.rs.std
        self.std = t.where(
            std < 1e-8, t.ones_like(std), std
        )

    def filter(self, x: t.Tensor, update: bool = True) -> t.Tensor:
        if update:
            for row in x.reshape(-1, *self.shape):
                self.buffer.push(row)
                self.rs.push(row)
        return (
            (x.to(t.float64) - self.mean) / self.std
        ).to(x.dtype)

    def __repr__(self):
        return f"MeanStdFilter(shape={self.shape})"


class SharedNoiseSampler:
    """Deterministic slices of a shared noise table."""

    def __init__(self, noise: t.Tensor, seed: int):
        self.noise = noise
        self.rng = np.random.RandomState(seed)

    def get(self, idx: int, size: int) -> t.Tensor:
        return self.noise[idx : idx + size]

    def sample(self, size: int) -> Tuple[int, t.Tensor]:
        idx = int(self.rng.randint(0, len(self.noise) - size + 1))
        return idx, self.get(idx, size)


class ARS(TorchFramework):
    _is_top = ["actor"]
    _is_restorable = ["actor"]

    def __init__(
        self,
        actor: nn.Module,
        optimizer,
        ars_group: RpcGroup,
        model_server: Tuple[PushPullModelServer],
        *_,
        lr_scheduler=None,
        lr_scheduler_args=(),
        lr_scheduler_kwargs=(),
        learning_rate: float = 0.01,
        gradient_max: float = np.inf,
        noise_std_dev: float = 0.02,
        noise_size: int = 25000000,
        rollout_num: int = 32,
        used_rollout_num: int = 32,
        normalize_state: bool = True,
        noise_seed: int = 12345,
        sample_seed: int = 123,
        **__,
    ):
        super().__init__()
        if used_rollout_num > rollout_num:
            raise ValueError("used_rollout_num must be <= rollout_num.")
        self.actor = actor
        self.ars_group = ars_group
        self.actor_model_server = model_server[0]
        self.grad_max = gradient_max
        self.noise_std_dev = noise_std_dev
        self.rollout_num = rollout_num
        self.used_rollout_num = used_rollout_num
        self.normalize_state = normalize_state

        self.actor_optim = optimizer(actor.parameters(), lr=learning_rate)
        self.actor_lr_sch = None
        if lr_scheduler is not None:
            args = lr_scheduler_args or ((),)
            kwargs = lr_scheduler_kwargs or ({},)
            self.actor_lr_sch = lr_scheduler(
                self.actor_optim, *args[0], **kwargs[0]
            )

        # same seed everywhere -> identical table on every process
        g = t.Generator().manual_seed(noise_seed)
        self.noise_table = t.randn(noise_size, generator=g,
                                   dtype=t.float32)
        me = ars_group.get_group_members().index(ars_group.get_cur_name())
        self.sampler = SharedNoiseSampler(
            self.noise_table, sample_seed + me
        )

        self._param_numel = sum(
            p.numel() for p in self.actor.parameters()
        )
        # rollout assignment by worker index (reference :332-339)
        members = ars_group.get_group_members()
        self._my_rollouts = [
            r
            for r in range(rollout_num)
            if r % len(members) == me
        ]
        self.filter: Dict[str, MeanStdFilter] = {}
        self._deltas: Dict[int, int] = {}     # rollout id -> noise idx
        self._rewards: Dict[str, float] = {}
        self._generate_deltas()

        # all members start from the first member's parameters
        self._sync_actor()

    # ------------------------------------------------------------------
    @property
    def optimizers(self):
        return [self.actor_optim]

    @optimizers.setter
    def optimizers(self, optimizers):
        self.actor_optim = optimizers[0]

    @property
    def lr_schedulers(self):
        return [self.actor_lr_sch] if self.actor_lr_sch else []

    @classmethod
    def is_distributed(cls) -> bool:
        return True

    # ------------------------------------------------------------------
    def _generate_deltas(self):
        self._deltas = {}
        self._rewards = {}
        for r in self._my_rollouts:
            idx, _ = self.sampler.sample(self._param_numel)
            self._deltas[r] = idx

    def get_actor_types(self) -> List[str]:
        """['original', 'pos_0', 'neg_0', ...] for assigned rollouts."""
        out = ["original"]
        for r in self._my_rollouts:
            out.append(f"pos_{r}")
            out.append(f"neg_{r}")
        return out

    def _apply_delta(self, sign: float, idx: int):
        noise = self.sampler.get(idx, self._param_numel)
        offset = 0
        with t.no_grad():
            for p in self.actor.parameters():
                n = p.numel()
                p.add_(
                    sign
                    * self.noise_std_dev
                    * noise[offset : offset + n].view_as(p).to(p.device)
                )
                offset += n

    def act(self, state: Dict[str, Any], actor_type: str, *_, **__):
        """Act with the original or a perturbed actor."""
        if actor_type != "original" and actor_type not in self.get_actor_types():
            raise ValueError(f"Invalid actor type {actor_type!r}")
        if self.normalize_state:
            for k, v in state.items():
                f = self.filter.get(k)
                if f is None:
                    f = MeanStdFilter(tuple(v.shape[1:]))
                    f.apply_stats()
                    self.filter[k] = f
                state = dict(state)
                state[k] = f.filter(v)
        if actor_type == "original":
            return safe_return(safe_call(self.actor, state))
        sign = 1.0 if actor_type.startswith("pos") else -1.0
        rollout = int(actor_type.split("_")[1])
        idx = self._deltas[rollout]
        self._apply_delta(sign, idx)
        try:
            result = safe_return(safe_call(self.actor, state))
        finally:
            self._apply_delta(-sign, idx)
        return result

    def store_reward(self, reward: float, actor_type: str, *_, **__):
        if actor_type not in self.get_actor_types():
            raise ValueError(f"Invalid actor type {actor_type!r}")
        if actor_type == "original":
            return
        self._rewards[actor_type] = self._rewards.get(actor_type, 0.0) + float(
            reward
        )

    # ------------------------------------------------------------------
    def update(self):
        """Exchange (reward+, reward-, noise idx) across the group,
        compute the ARS gradient on every member, step, resample."""
        me = self.ars_group.get_cur_name()
        local = {
            r: (
                self._rewards.get(f"pos_{r}", 0.0),
                self._rewards.get(f"neg_{r}", 0.0),
                self._deltas[r],
            )
            for r in self._my_rollouts
        }
        self.ars_group.pair(f"ars_rollouts_{me}", local)
        # exchange filters too
        if self.normalize_state:
            self.ars_group.pair(
                f"ars_filter_{me}",
                {k: f.copy() for k, f in self.filter.items()},
            )
        self.ars_group.barrier()

        all_rollouts = {}
        for m in self.ars_group.get_group_members():
            all_rollouts.update(
                self.ars_group.get_paired(f"ars_rollouts_{m}").to_here()
            )
        if self.normalize_state:
            for m in self.ars_group.get_group_members():
                if m == me:
                    continue
                other = self.ars_group.get_paired(
                    f"ars_filter_{m}"
                ).to_here()
                for k, f in other.items():
                    if k in self.filter:
                        self.filter[k].collect(f)
                    else:
                        self.filter[k] = f.copy()
            for f in self.filter.values():
                f.apply_stats()
                f.clear_local()
        self.ars_group.barrier()
        # cleanup pairs for the next round
        self.ars_group.unpair(f"ars_rollouts_{me}")
        if self.normalize_state:
            self.ars_group.unpair(f"ars_filter_{me}")
        self.ars_group.barrier()

        # top-percentile direction selection by max(r+, r-)
        items = sorted(
            all_rollouts.items(),
            key=lambda kv: max(kv[1][0], kv[1][1]),
            reverse=True,
        )[: self.used_rollout_num]
        rewards = np.array(
            [[rp, rn] for _, (rp, rn, _) in items], dtype=np.float64
        )
        reward_std = rewards.std() if rewards.size else 1.0
        if reward_std < 1e-8:
            reward_std = 1.0

        grad = t.zeros(self._param_numel, dtype=t.float32)
        for _, (rp, rn, idx) in items:
            grad += (rp - rn) * self.sampler.get(idx, self._param_numel)
        grad /= len(items) * reward_std

        # apply as a gradient ascent step through the optimizer
        offset = 0
        for p in self.actor.parameters():
            n = p.numel()
            p.grad = (-grad[offset : offset + n]).view_as(p).to(p.device)
            offset += n
        nn.utils.clip_grad_norm_(self.actor.parameters(), self.grad_max)
        self.actor_optim.step()
        self.actor_optim.zero_grad(set_to_none=False)

        # drift guard: re-sync everyone from the first member
        self._sync_actor()
        self._generate_deltas()
        return True

    def _sync_actor(self):
        members = self.ars_group.get_group_members()
        me = self.ars_group.get_cur_name()
        if me == members[0]:
            self.actor_model_server.push(self.actor)
        self.ars_group.barrier()
        if me != members[0]:
            self.actor_model_server.pull(self.actor)
        self.ars_group.barrier()

    def update_lr_scheduler(self):
        if self.actor_lr_sch is not None:
            self.actor_lr_sch.step()
