"""TD3: twin-delayed DDPG.

Parity target: reference ``machin/frame/algorithms/td3.py`` (:117-300):
DDPG + second critic pair, target-policy smoothing via
``policy_noise_function``, min-of-two target values, delayed policy
updates.
"""
from typing import Callable, Union

import torch as t
import torch.nn as nn

from .ddpg import DDPG
from .utils import hard_update, safe_call, safe_return, soft_update


class TD3(DDPG):
    _is_top = [
        "actor", "actor_target", "critic", "critic_target", "critic2",
        "critic2_target",
    ]
    _is_restorable = ["actor_target", "critic_target", "critic2_target"]

    def __init__(
        self,
        actor: nn.Module,
        actor_target: nn.Module,
        critic: nn.Module,
        critic_target: nn.Module,
        critic2: nn.Module,
        critic2_target: nn.Module,
        optimizer: Callable,
        criterion: Callable,
        *_,
        policy_update_delay: int = 2,
        critic_learning_rate: float = 0.001,
        **kwargs,
    ):
        super().__init__(
            actor, actor_target, critic, critic_target, optimizer, criterion,
            critic_learning_rate=critic_learning_rate, **kwargs,
        )
        self.critic2 = critic2
        self.critic2_target = critic2_target
        optim_cls = (
            optimizer[1] if isinstance(optimizer, (tuple, list)) else optimizer
        )
        self.critic2_optim = optim_cls(
            self.critic2.parameters(), lr=critic_learning_rate
        )
        hard_update(self.critic2_target, self.critic2)
        self.policy_update_delay = policy_update_delay
        self._policy_update_counter = 0

    @property
    def optimizers(self):
        return [self.actor_optim, self.critic_optim, self.critic2_optim]

    @optimizers.setter
    def optimizers(self, optimizers):
        self.actor_optim, self.critic_optim, self.critic2_optim = optimizers

    def _criticize2(self, state, action, use_target=False, **__):
        net = self.critic2_target if use_target else self.critic2
        return safe_return(safe_call(net, state, action))

    def update(
        self,
        update_value=True,
        update_policy=True,
        update_target=True,
        concatenate_samples=True,
        **__,
    ):
        (
            batch_size,
            (state, action, reward, next_state, terminal, others),
        ) = self.replay_buffer.sample_batch(
            self.batch_size,
            concatenate_samples,
            sample_method="random_unique",
            sample_attrs=["state", "action", "reward", "next_state", "terminal", "*"],
        )
        if batch_size == 0:
            return 0.0, 0.0
        self.actor.train()
        self.critic.train()
        self.critic2.train()

        with t.no_grad():
            raw_next = self.act(next_state, True)
            raw_next = self.policy_noise_function(raw_next)
            next_action = self.action_transform_function(
                raw_next, next_state, others
            )
            nv1 = self._criticize(next_state, next_action, True)
            nv2 = self._criticize2(next_state, next_action, True)
            next_value = t.min(nv1, nv2)
            device = next_value.device
            reward = reward.to(device).float().view(batch_size, 1)
            terminal = terminal.to(device).float().view(batch_size, 1)
            y = self.reward_function(
                reward, self.discount, next_value.view(batch_size, 1),
                terminal, others,
            )

        cur_v1 = self._criticize(state, action)
        cur_v2 = self._criticize2(state, action)
        value_loss = self.criterion(cur_v1, y.to(cur_v1.dtype))
        value_loss2 = self.criterion(cur_v2, y.to(cur_v2.dtype))
        if self.visualize:
            self.visualize_model(value_loss, "critic", self.visualize_dir)
        if update_value:
            self.critic_optim.zero_grad(set_to_none=True)
            self._backward(value_loss)
            nn.utils.clip_grad_norm_(self.critic.parameters(), self.grad_max)
            self.critic_optim.step()
            self.critic2_optim.zero_grad(set_to_none=True)
            self._backward(value_loss2)
            nn.utils.clip_grad_norm_(self.critic2.parameters(), self.grad_max)
            self.critic2_optim.step()

        act_policy_loss = t.zeros(1)
        self._policy_update_counter += 1
        do_policy = (
            self._policy_update_counter % self.policy_update_delay == 0
        )
        if do_policy:
            cur_action = self.action_transform_function(
                self.act(state), state, others
            )
            act_value = self._criticize(state, cur_action)
            act_policy_loss = -act_value.mean()
            if self.visualize:
                self.visualize_model(
                    act_policy_loss, "actor", self.visualize_dir
                )
            if update_policy:
                self.actor_optim.zero_grad(set_to_none=True)
                self._backward(act_policy_loss)
                nn.utils.clip_grad_norm_(
                    self.actor.parameters(), self.grad_max
                )
                self.actor_optim.step()
            if update_target:
                if self.update_rate is not None:
                    soft_update(self.actor_target, self.actor, self.update_rate)
                    soft_update(
                        self.critic_target, self.critic, self.update_rate
                    )
                    soft_update(
                        self.critic2_target, self.critic2, self.update_rate
                    )
                else:
                    self._update_counter += 1
                    if self._update_counter % self.update_steps == 0:
                        hard_update(self.actor_target, self.actor)
                        hard_update(self.critic_target, self.critic)
                        hard_update(self.critic2_target, self.critic2)

        return (
            -float(act_policy_loss.detach().item()),
            float(
                ((value_loss + value_loss2) / 2).detach().item()
            ),
        )

    def load(self, model_dir, network_map=None, version=-1):
        # skip DDPG.load (it only re-syncs actor/critic): restore all
        # three target nets, then hard-update the online nets
        from .base import TorchFramework

        TorchFramework.load(self, model_dir, network_map, version)
        with t.no_grad():
            hard_update(self.actor, self.actor_target)
            hard_update(self.critic, self.critic_target)
            hard_update(self.critic2, self.critic2_target)

    @staticmethod
    def policy_noise_function(actions: t.Tensor, *_) -> t.Tensor:
        """Target-policy smoothing: clipped gaussian noise on the
        target action (override for custom bounds)."""
        noise = (t.randn_like(actions) * 0.2).clamp(-0.5, 0.5)
        return actions + noise

    @classmethod
    def generate_config(cls, config):
        config = DDPG.generate_config(config)
        fc = config["frame_config"]
        fc["frame"] = "TD3"
        fc.setdefault("models", ["Actor", "Actor", "Critic", "Critic",
                                 "Critic", "Critic"])
        if len(fc["models"]) == 4:
            fc["models"] = fc["models"] + [fc["models"][2], fc["models"][3]]
        fc.setdefault("policy_update_delay", 2)
        while len(fc["model_args"]) < 6:
            fc["model_args"] = tuple(fc["model_args"]) + (
                fc["model_args"][-1],
            )
        while len(fc["model_kwargs"]) < 6:
            fc["model_kwargs"] = tuple(fc["model_kwargs"]) + (
                fc["model_kwargs"][-1],
            )
        config["frame"] = "TD3"
        return config
