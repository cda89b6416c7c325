"""PPO: clipped-surrogate policy optimization on top of A2C.

Parity target: reference ``machin/frame/algorithms/ppo.py``
(:111-169): old log-probs frozen at the start of ``update`` (the
reference deepcopies the pre-update actor — here we evaluate them
once under no_grad before the first gradient step, which is the same
quantity without the copy), ratio clipped to ``1 ± surrogate_loss_clip``,
min of clipped/unclipped surrogates.
"""
import torch as t
import torch.nn as nn

from .a2c import A2C


class PPO(A2C):
    def __init__(
        self,
        actor: nn.Module,
        critic: nn.Module,
        optimizer,
        criterion,
        *_,
        surrogate_loss_clip: float = 0.2,
        **kwargs,
    ):
        super().__init__(actor, critic, optimizer, criterion, **kwargs)
        self.surr_clip = surrogate_loss_clip

    def update(self, update_value=True, update_policy=True,
               concatenate_samples=True, **__):
        (
            batch_size,
            (state, action, reward, next_state, terminal, target_value,
             advantage),
        ) = self.replay_buffer.sample_batch(
            -1,
            sample_method="all",
            concatenate=concatenate_samples,
            sample_attrs=[
                "state", "action", "reward", "next_state", "terminal",
                "value", "gae",
            ],
            additional_concat_custom_attrs=["value", "gae"],
        )
        if batch_size == 0:
            return 0.0, 0.0
        self.actor.train()
        self.critic.train()

        # freeze old log-probs (pre-update policy)
        with t.no_grad():
            old_result = self._eval_act(state, action)
            old_log_prob = old_result[1].view(batch_size, 1)

        sum_act_loss = 0.0
        sum_value_loss = 0.0

        for _ in range(self.actor_update_times):
            result = self._eval_act(state, action)
            new_log_prob = result[1].view(batch_size, 1)
            entropy = (
                result[2].view(batch_size, 1) if len(result) > 2 else None
            )
            adv = advantage.to(new_log_prob.device).view(batch_size, 1).detach()
            if self.normalize_advantage:
                adv = (adv - adv.mean()) / (adv.std() + 1e-6)
            ratio = (new_log_prob - old_log_prob.to(new_log_prob.device)).exp()
            surr1 = ratio * adv
            surr2 = ratio.clamp(
                1.0 - self.surr_clip, 1.0 + self.surr_clip
            ) * adv
            act_policy_loss = -t.min(surr1, surr2)
            if self.entropy_weight is not None and entropy is not None:
                act_policy_loss += self.entropy_weight * -entropy
            act_policy_loss = act_policy_loss.mean()
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
            sum_act_loss += float(act_policy_loss.detach().item())

        for _ in range(self.critic_update_times):
            value = self._criticize(state).view(batch_size, 1)
            tv = target_value.to(value.device).view(batch_size, 1)
            value_loss = (
                self.criterion(value, tv.to(value.dtype)) * self.value_weight
            )
            if self.visualize:
                self.visualize_model(value_loss, "critic", self.visualize_dir)
            if update_value:
                self.critic_optim.zero_grad(set_to_none=True)
                self._backward(value_loss)
                nn.utils.clip_grad_norm_(
                    self.critic.parameters(), self.grad_max
                )
                self.critic_optim.step()
            sum_value_loss += float(value_loss.detach().item())

        self.replay_buffer.clear()
        return (
            -sum_act_loss / max(self.actor_update_times, 1),
            sum_value_loss / max(self.critic_update_times, 1),
        )

    @classmethod
    def generate_config(cls, config):
        config = A2C.generate_config(config)
        config["frame"] = "PPO"
        config["frame_config"]["frame"] = "PPO"
        config["frame_config"].setdefault("surrogate_loss_clip", 0.2)
        return config
