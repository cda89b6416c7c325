"""DDPG with prioritized experience replay.

Parity target: reference ``machin/frame/algorithms/ddpg_per.py``:
IS-weighted critic loss + abs-TD priority update on a
PrioritizedBuffer.
"""
from typing import Union

import numpy as np
import torch as t
import torch.nn as nn

from ..buffers import default_buffer as _default_buffer
from ..buffers.prioritized_buffer import PrioritizedBuffer
from .ddpg import DDPG
from .utils import hard_update, soft_update


class DDPGPer(DDPG):
    def __init__(
        self,
        actor,
        actor_target,
        critic,
        critic_target,
        optimizer,
        criterion,
        *_,
        replay_size: int = 500000,
        replay_device: Union[str, t.device] = "cpu",
        replay_buffer=None,
        **kwargs,
    ):
        super().__init__(
            actor,
            actor_target,
            critic,
            critic_target,
            optimizer,
            criterion,
            replay_buffer=(
                # cuda replay_device -> DeviceSumTree PER in HBM
                _default_buffer(replay_size, replay_device,
                                prioritized=True)
                if replay_buffer is None
                else replay_buffer
            ),
            **kwargs,
        )
        crit = self.criterion
        if getattr(crit, "reduction", None) not in (None, "none"):
            crit.reduction = "none"

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
            index,
            is_weight,
        ) = self.replay_buffer.sample_batch(
            self.batch_size,
            concatenate_samples,
            sample_attrs=["state", "action", "reward", "next_state", "terminal", "*"],
        )
        if batch_size == 0:
            return 0.0, 0.0
        self.actor.train()
        self.critic.train()

        with t.no_grad():
            next_action = self.action_transform_function(
                self.act(next_state, True), next_state, others
            )
            next_value = self._criticize(next_state, next_action, True)
            device = next_value.device
            reward = reward.to(device).float().view(batch_size, 1)
            terminal = terminal.to(device).float().view(batch_size, 1)
            y = self.reward_function(
                reward, self.discount, next_value.view(batch_size, 1),
                terminal, others,
            )

        cur_value = self._criticize(state, action)
        per_sample = self.criterion(cur_value, y.to(cur_value.dtype)).view(
            batch_size, -1
        ).sum(dim=1)
        weights = t.as_tensor(
            is_weight, dtype=per_sample.dtype, device=per_sample.device
        ).view(batch_size)
        value_loss = (per_sample * weights).mean()

        abs_td = (cur_value - y).detach().abs().view(batch_size)
        if getattr(self.replay_buffer, "accepts_tensor_priorities", False):
            self.replay_buffer.update_priority(abs_td, index)
        else:
            self.replay_buffer.update_priority(
                abs_td.cpu().numpy().astype(np.float64), index
            )

        if self.visualize:
            self.visualize_model(value_loss, "critic", self.visualize_dir)
        if update_value:
            self.critic_optim.zero_grad(set_to_none=True)
            self._backward(value_loss)
            nn.utils.clip_grad_norm_(self.critic.parameters(), self.grad_max)
            self.critic_optim.step()

        cur_action = self.action_transform_function(
            self.act(state), state, others
        )
        act_value = self._criticize(state, cur_action)
        act_policy_loss = -act_value.mean()
        if self.visualize:
            self.visualize_model(act_policy_loss, "actor", self.visualize_dir)
        if update_policy:
            self.actor_optim.zero_grad(set_to_none=True)
            self._backward(act_policy_loss)
            nn.utils.clip_grad_norm_(self.actor.parameters(), self.grad_max)
            self.actor_optim.step()

        if update_target:
            if self.update_rate is not None:
                soft_update(self.actor_target, self.actor, self.update_rate)
                soft_update(self.critic_target, self.critic, self.update_rate)
            else:
                self._update_counter += 1
                if self._update_counter % self.update_steps == 0:
                    hard_update(self.actor_target, self.actor)
                    hard_update(self.critic_target, self.critic)

        return (
            -float(act_policy_loss.detach().item()),
            float(value_loss.detach().item()),
        )
