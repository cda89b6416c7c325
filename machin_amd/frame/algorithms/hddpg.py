"""Hysteretic DDPG.

Parity target: reference ``machin/frame/algorithms/hddpg.py``
(:29-137): DDPG with asymmetric TD-error scaling — increases scaled by
``q_increase_rate``, decreases by ``q_decrease_rate``.
"""
import torch as t
import torch.nn as nn

from .ddpg import DDPG


class HDDPG(DDPG):
    def __init__(
        self,
        actor,
        actor_target,
        critic,
        critic_target,
        optimizer,
        criterion,
        *_,
        q_increase_rate: float = 1.0,
        q_decrease_rate: float = 1.0,
        **kwargs,
    ):
        super().__init__(
            actor, actor_target, critic, critic_target, optimizer, criterion,
            **kwargs,
        )
        self.q_increase_rate = q_increase_rate
        self.q_decrease_rate = q_decrease_rate

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
        # hysteretic scaling of the TD error
        with t.no_grad():
            delta = y - cur_value
            scale = t.where(
                delta > 0,
                t.full_like(delta, self.q_increase_rate),
                t.full_like(delta, self.q_decrease_rate),
            )
            y_scaled = cur_value + delta * scale
        value_loss = self.criterion(cur_value, y_scaled.to(cur_value.dtype))
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
            from .utils import hard_update, soft_update

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
