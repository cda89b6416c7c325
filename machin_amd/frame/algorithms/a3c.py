"""A3C: asynchronous advantage actor-critic via gradient servers.

Parity target: reference ``machin/frame/algorithms/a3c.py`` (:87-165):
A2C with a no-op local optimizer (FakeOptimizer); every ``act`` /
``_criticize`` pulls fresh parameters from the gradient servers while
``is_syncing``; ``update()`` runs the local A2C backward then pushes
gradients to the actor/critic grad servers (which step the real
optimizer remotely).
"""
from typing import Tuple

import torch.nn as nn

from ...parallel.server.param_server import PushPullGradServer
from .a2c import A2C
from .utils import FakeOptimizer


class A3C(A2C):
    def __init__(
        self,
        actor: nn.Module,
        critic: nn.Module,
        criterion,
        grad_server: Tuple[PushPullGradServer, PushPullGradServer],
        *_,
        batch_size: int = 100,
        actor_update_times: int = 5,
        critic_update_times: int = 10,
        entropy_weight: float = None,
        value_weight: float = 0.5,
        gradient_max: float = None,
        gae_lambda: float = 1.0,
        discount: float = 0.99,
        normalize_advantage: bool = True,
        replay_size: int = 500000,
        replay_device="cpu",
        replay_buffer=None,
        visualize: bool = False,
        visualize_dir: str = "",
        **__,
    ):
        import numpy as np

        super().__init__(
            actor,
            critic,
            FakeOptimizer,
            criterion,
            batch_size=batch_size,
            actor_update_times=actor_update_times,
            critic_update_times=critic_update_times,
            entropy_weight=entropy_weight,
            value_weight=value_weight,
            gradient_max=(
                gradient_max if gradient_max is not None else np.inf
            ),
            gae_lambda=gae_lambda,
            discount=discount,
            normalize_advantage=normalize_advantage,
            replay_size=replay_size,
            replay_device=replay_device,
            replay_buffer=replay_buffer,
            visualize=visualize,
            visualize_dir=visualize_dir,
        )
        self.actor_grad_server, self.critic_grad_server = (
            grad_server[0],
            grad_server[1],
        )
        self.is_syncing = True

    @classmethod
    def is_distributed(cls) -> bool:
        return True

    def set_sync(self, is_syncing: bool):
        self.is_syncing = is_syncing

    def manual_sync(self):
        self.actor_grad_server.pull(self.actor)
        self.critic_grad_server.pull(self.critic)

    def act(self, state, *_, **__):
        if self.is_syncing:
            self.actor_grad_server.pull(self.actor)
        return super().act(state)

    def _eval_act(self, state, action, **__):
        if self.is_syncing:
            self.actor_grad_server.pull(self.actor)
        return super()._eval_act(state, action)

    def _criticize(self, state, *_, **__):
        if self.is_syncing:
            self.critic_grad_server.pull(self.critic)
        return super()._criticize(state)

    def update(self, update_value=True, update_policy=True,
               concatenate_samples=True, **__):
        """Local A2C backward (optimizer is a no-op), then push
        gradients to the servers and pull fresh parameters."""
        org_sync = self.is_syncing
        self.is_syncing = False
        result = super().update(update_value, update_policy,
                                concatenate_samples)
        self.is_syncing = org_sync
        if update_policy:
            self.actor_grad_server.push(self.actor)
        if update_value:
            self.critic_grad_server.push(self.critic)
        return result

    @classmethod
    def generate_config(cls, config):
        config = A2C.generate_config(config)
        config["frame"] = "A3C"
        fc = config["frame_config"]
        fc["frame"] = "A3C"
        fc.setdefault("grad_server_group_name", "grad_server_group")
        fc.setdefault("grad_server_members", "all")
        return config

    @classmethod
    def init_from_config(cls, config, model_device="cpu"):
        from ...frame.helpers.servers import grad_server_helper
        from .utils import (
            assert_and_get_valid_criterion,
            assert_and_get_valid_models,
        )

        data = config.data if hasattr(config, "data") else dict(config)
        fc = data["frame_config"]
        model_cls = assert_and_get_valid_models(fc["models"])
        model_args = fc.get("model_args", ((), ()))
        model_kwargs = fc.get("model_kwargs", ({}, {}))
        models = [
            m(*args, **kwargs).to(model_device)
            for m, args, kwargs in zip(model_cls, model_args, model_kwargs)
        ]
        creators = [
            (lambda mi=m, a=args, k=kwargs: mi(*a, **k))
            for m, args, kwargs in zip(model_cls, model_args, model_kwargs)
        ]
        servers = grad_server_helper(
            creators,
            learning_rate=fc.get("learning_rate", 1e-3),
        )
        criterion = assert_and_get_valid_criterion(fc["criterion"])(
            *fc.get("criterion_args", ()), **fc.get("criterion_kwargs", {})
        )
        return cls(
            models[0],
            models[1],
            criterion,
            servers,
            **{
                k: v
                for k, v in fc.items()
                if k
                not in (
                    "frame", "models", "model_args", "model_kwargs",
                    "optimizer", "criterion", "criterion_args",
                    "criterion_kwargs", "lr_scheduler",
                    "grad_server_group_name", "grad_server_members",
                )
            },
        )
