"""APEX: distributed prioritized experience replay (DQN and DDPG).

Parity target: reference ``machin/frame/algorithms/apex.py``:
``DQNApex`` (:105-253) and ``DDPGApex`` (:356-532) — replay replaced
by a :class:`DistributedPrioritizedBuffer` on the apex RpcGroup;
sampler processes pull fresh models from a ``PushPullModelServer``
before acting (:123-139); the learner pushes after every ``update()``
(:141-150).

MI355X note (``init_from_config``): learner ranks wrap their models in
machin_amd.parallel.ddp.DistributedDataParallel — bucketed RCCL
all-reduce over xGMI on a collective group of the first
``learner_process_number`` ranks (reference wraps torch DDP at
apex.py:213-221).
"""
from typing import Tuple

import torch as t
import torch.nn as nn

from ...parallel.distributed.world import RpcGroup
from ...parallel.server.param_server import PushPullModelServer
from ..buffers.prioritized_buffer_d import DistributedPrioritizedBuffer
from .ddpg_per import DDPGPer
from .dqn_per import DQNPer


def _resolve_device_replay(fc, world, learner_n, model_device):
    """When the config asks for cuda replay, resolve the per-rank
    shard device (each learner rank owns its own GPU) and name the
    learner members that host HBM shards."""
    rd = fc.get("replay_device", "cpu")
    if t.device(rd).type == "cuda":
        if t.device(model_device).type == "cuda":
            fc["replay_device"] = model_device
        fc["replay_learners"] = world.get_members()[:learner_n]
    return fc


def _apex_buffer(name, group, replay_size, replay_device,
                 replay_learners):
    """Pick the replay backend: learner-resident HBM shards
    (store-side push, sampling local to the learner GPU) when the
    replay device is cuda or learner shards are named; otherwise the
    reference-style sample-side-pull CPU buffer."""
    if t.device(replay_device).type == "cuda" or replay_learners:
        from ..buffers.device_buffer_d import (
            DeviceDistributedPrioritizedBuffer,
        )

        return DeviceDistributedPrioritizedBuffer(
            name, group, replay_size, learners=replay_learners,
            device=replay_device,
        )
    return DistributedPrioritizedBuffer(name, group, replay_size)


class DQNApex(DQNPer):
    def __init__(
        self,
        qnet: nn.Module,
        qnet_target: nn.Module,
        optimizer,
        criterion,
        apex_group: RpcGroup,
        model_server: Tuple[PushPullModelServer],
        *_,
        replay_size: int = 500000,
        replay_device="cpu",
        replay_learners=None,
        **kwargs,
    ):
        buffer = _apex_buffer(
            "dqn_apex_buffer", apex_group, replay_size, replay_device,
            replay_learners,
        )
        super().__init__(
            qnet, qnet_target, optimizer, criterion,
            replay_buffer=buffer, **kwargs,
        )
        self.apex_group = apex_group
        self.qnet_model_server = model_server[0]
        self.is_syncing = True

    @classmethod
    def is_distributed(cls) -> bool:
        return True

    def set_sync(self, is_syncing: bool):
        self.is_syncing = is_syncing

    def manual_sync(self):
        self.qnet_model_server.pull(self.qnet)

    def act_discrete_with_noise(self, state, use_target=False,
                                decay_epsilon=True, **__):
        if self.is_syncing:
            self.qnet_model_server.pull(self.qnet)
        return super().act_discrete_with_noise(
            state, use_target, decay_epsilon
        )

    def act_discrete(self, state, use_target=False, **__):
        if self.is_syncing:
            self.qnet_model_server.pull(self.qnet)
        return super().act_discrete(state, use_target)

    def update(self, update_value=True, update_target=True,
               concatenate_samples=True, **__):
        result = super().update(update_value, update_target,
                                concatenate_samples)
        if update_value:
            self.qnet_model_server.push(
                getattr(self.qnet, "module", self.qnet)
            )
        return result

    @classmethod
    def generate_config(cls, config):
        config = DQNPer.generate_config(config)
        config["frame"] = "DQNApex"
        fc = config["frame_config"]
        fc["frame"] = "DQNApex"
        fc.setdefault("apex_group_name", "apex_group")
        fc.setdefault("apex_group_members", "all")
        fc.setdefault("learner_process_number", 1)
        return config

    @classmethod
    def init_from_config(cls, config, model_device="cpu"):
        from ...frame.helpers.servers import model_server_helper
        from ...parallel.ddp import DistributedDataParallel
        from ...parallel.distributed.world import get_world
        from .utils import (
            assert_and_get_valid_criterion,
            assert_and_get_valid_models,
            assert_and_get_valid_optimizer,
        )

        data = config.data if hasattr(config, "data") else dict(config)
        fc = data["frame_config"]
        world = get_world()
        servers = model_server_helper(model_num=1)
        members = (
            world.get_members()
            if fc.get("apex_group_members", "all") == "all"
            else fc["apex_group_members"]
        )
        group = world.create_rpc_group(
            fc.get("apex_group_name", "apex_group"), members
        )
        model_cls = assert_and_get_valid_models(fc["models"])
        models = [
            m(*args, **kwargs).to(model_device)
            for m, args, kwargs in zip(
                model_cls, fc.get("model_args", ((), ())),
                fc.get("model_kwargs", ({}, {})),
            )
        ]
        learner_n = fc.get("learner_process_number", 1)
        learner_ranks = list(range(learner_n))
        fc = _resolve_device_replay(
            dict(fc), world, learner_n, model_device
        )
        # every process must join new_group creation collectively
        coll = world.create_collective_group(learner_ranks) \
            if world.rank in learner_ranks and learner_n > 1 else None
        if world.rank in learner_ranks and learner_n > 1:
            models[0] = DistributedDataParallel(
                models[0], process_group=coll.group,
                reduction=fc.get("ddp_reduction", "all_reduce"),
            )
        optimizer = assert_and_get_valid_optimizer(fc["optimizer"])
        criterion = assert_and_get_valid_criterion(fc["criterion"])(
            *fc.get("criterion_args", ()), **fc.get("criterion_kwargs", {})
        )
        frame = cls(
            models[0], models[1], optimizer, criterion, group, servers,
            **{
                k: v
                for k, v in fc.items()
                if k
                not in (
                    "frame", "models", "model_args", "model_kwargs",
                    "optimizer", "criterion", "criterion_args",
                    "criterion_kwargs", "lr_scheduler",
                    "apex_group_name", "apex_group_members",
                    "learner_process_number", "ddp_reduction",
                )
            },
        )
        if world.rank not in learner_ranks:
            # sampler ranks never update
            frame.update = lambda *a, **k: 0.0
        else:
            # wire reducer finalize + bucket-preserving zero_grad into
            # the framework's pluggable backward (no-op when no model
            # is DDP-wrapped)
            from ...parallel.ddp import install_ddp_finalize

            install_ddp_finalize(frame)
        return frame


class DDPGApex(DDPGPer):
    def __init__(
        self,
        actor: nn.Module,
        actor_target: nn.Module,
        critic: nn.Module,
        critic_target: nn.Module,
        optimizer,
        criterion,
        apex_group: RpcGroup,
        model_server: Tuple[PushPullModelServer],
        *_,
        replay_size: int = 500000,
        replay_device="cpu",
        replay_learners=None,
        **kwargs,
    ):
        buffer = _apex_buffer(
            "ddpg_apex_buffer", apex_group, replay_size, replay_device,
            replay_learners,
        )
        super().__init__(
            actor, actor_target, critic, critic_target, optimizer,
            criterion, replay_buffer=buffer, **kwargs,
        )
        self.apex_group = apex_group
        self.actor_model_server = model_server[0]
        self.is_syncing = True

    @classmethod
    def is_distributed(cls) -> bool:
        return True

    def set_sync(self, is_syncing: bool):
        self.is_syncing = is_syncing

    def manual_sync(self):
        self.actor_model_server.pull(self.actor)

    def act(self, state, use_target=False, **__):
        if self.is_syncing and not use_target:
            self.actor_model_server.pull(self.actor)
        return super().act(state, use_target)

    def act_with_noise(self, state, noise_param=(0.0, 1.0), ratio=1.0,
                       mode="uniform", use_target=False, **__):
        if self.is_syncing and not use_target:
            self.actor_model_server.pull(self.actor)
        org = self.is_syncing
        self.is_syncing = False
        try:
            return super().act_with_noise(
                state, noise_param, ratio, mode, use_target
            )
        finally:
            self.is_syncing = org

    def act_discrete(self, state, use_target=False, **__):
        if self.is_syncing and not use_target:
            self.actor_model_server.pull(self.actor)
        org = self.is_syncing
        self.is_syncing = False
        try:
            return super().act_discrete(state, use_target)
        finally:
            self.is_syncing = org

    def act_discrete_with_noise(self, state, use_target=False, **__):
        if self.is_syncing and not use_target:
            self.actor_model_server.pull(self.actor)
        org = self.is_syncing
        self.is_syncing = False
        try:
            return super().act_discrete_with_noise(state, use_target)
        finally:
            self.is_syncing = org

    def update(self, update_value=True, update_policy=True,
               update_target=True, concatenate_samples=True, **__):
        org = self.is_syncing
        self.is_syncing = False
        try:
            result = super().update(
                update_value, update_policy, update_target,
                concatenate_samples,
            )
        finally:
            self.is_syncing = org
        if update_policy:
            self.actor_model_server.push(
                getattr(self.actor, "module", self.actor)
            )
        return result

    @classmethod
    def generate_config(cls, config):
        config = DDPGPer.generate_config(config)
        config["frame"] = "DDPGApex"
        fc = config["frame_config"]
        fc["frame"] = "DDPGApex"
        fc.setdefault("apex_group_name", "apex_group")
        fc.setdefault("apex_group_members", "all")
        fc.setdefault("learner_process_number", 1)
        return config

    @classmethod
    def init_from_config(cls, config, model_device="cpu"):
        from ...frame.helpers.servers import model_server_helper
        from ...parallel.ddp import (
            DistributedDataParallel,
            install_ddp_finalize,
        )
        from ...parallel.distributed.world import get_world
        from .utils import (
            assert_and_get_valid_criterion,
            assert_and_get_valid_models,
            assert_and_get_valid_optimizer,
        )

        data = config.data if hasattr(config, "data") else dict(config)
        fc = data["frame_config"]
        world = get_world()
        servers = model_server_helper(model_num=1)
        members = (
            world.get_members()
            if fc.get("apex_group_members", "all") == "all"
            else fc["apex_group_members"]
        )
        group = world.create_rpc_group(
            fc.get("apex_group_name", "apex_group"), members
        )
        model_cls = assert_and_get_valid_models(fc["models"])
        models = [
            m(*args, **kwargs).to(model_device)
            for m, args, kwargs in zip(
                model_cls, fc.get("model_args", ((), (), (), ())),
                fc.get("model_kwargs", ({}, {}, {}, {})),
            )
        ]
        learner_n = fc.get("learner_process_number", 1)
        learner_ranks = list(range(learner_n))
        fc = _resolve_device_replay(
            dict(fc), world, learner_n, model_device
        )
        coll = world.create_collective_group(learner_ranks) \
            if world.rank in learner_ranks and learner_n > 1 else None
        if coll is not None:
            # learner ranks sync actor (models[0]) and critic
            # (models[2]) gradients; targets follow via soft_update
            reduction = fc.get("ddp_reduction", "all_reduce")
            models[0] = DistributedDataParallel(
                models[0], process_group=coll.group, reduction=reduction
            )
            models[2] = DistributedDataParallel(
                models[2], process_group=coll.group, reduction=reduction
            )
        optimizer = assert_and_get_valid_optimizer(fc["optimizer"])
        criterion = assert_and_get_valid_criterion(fc["criterion"])(
            *fc.get("criterion_args", ()), **fc.get("criterion_kwargs", {})
        )
        frame = cls(
            models[0], models[1], models[2], models[3],
            optimizer, criterion, group, servers,
            **{
                k: v
                for k, v in fc.items()
                if k
                not in (
                    "frame", "models", "model_args", "model_kwargs",
                    "optimizer", "criterion", "criterion_args",
                    "criterion_kwargs", "lr_scheduler",
                    "apex_group_name", "apex_group_members",
                    "learner_process_number", "ddp_reduction",
                )
            },
        )
        if world.rank not in learner_ranks:
            frame.update = lambda *a, **k: (0.0, 0.0)
        else:
            install_ddp_finalize(frame)
        return frame
