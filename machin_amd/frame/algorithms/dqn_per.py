"""DQN with prioritized experience replay.

Parity target: reference ``machin/frame/algorithms/dqn_per.py``
(:105-139): double-DQN target, importance-sampling-weighted loss
(criterion forced to reduction="none"), abs-TD priority write-back.
"""
from typing import Union

import numpy as np
import torch as t
import torch.nn as nn

from ..buffers.prioritized_buffer import PrioritizedBuffer
from .dqn import DQN


class DQNPer(DQN):
    def __init__(
        self,
        qnet,
        qnet_target,
        optimizer,
        criterion,
        *_,
        replay_size: int = 500000,
        replay_device: Union[str, t.device] = "cpu",
        replay_buffer=None,
        **kwargs,
    ):
        kwargs.pop("mode", None)
        from ..buffers import default_buffer

        super().__init__(
            qnet,
            qnet_target,
            optimizer,
            criterion,
            replay_buffer=(
                # cuda replay_device -> DeviceSumTree PER in HBM
                default_buffer(replay_size, replay_device,
                               prioritized=True)
                if replay_buffer is None
                else replay_buffer
            ),
            mode="double",
            **kwargs,
        )
        # per-sample losses are needed to weight by IS weights
        crit = self.criterion
        if getattr(crit, "reduction", None) not in (None, "none"):
            crit.reduction = "none"

    def update(
        self, update_value=True, update_target=True, concatenate_samples=True,
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
            return 0.0
        self.qnet.train()

        with t.no_grad():
            online_next = self._criticize(next_state)
            best_action = online_next.argmax(dim=1, keepdim=True)
            q_next = self._criticize(next_state, use_target=True)
            target = q_next.gather(dim=1, index=best_action.to(q_next.device))
            device = target.device
            reward = reward.to(device).float().view(batch_size, 1)
            terminal = terminal.to(device).float().view(batch_size, 1)
            y = reward + self.discount * (1.0 - terminal) * target

        q = self._criticize(state)
        action_index = self._sampled_action_index(action, q)
        q_taken = q.gather(dim=1, index=action_index)
        per_sample = self.criterion(q_taken, y.to(q_taken.dtype)).view(
            batch_size, -1
        ).sum(dim=1)
        weights = t.as_tensor(
            is_weight, dtype=per_sample.dtype, device=per_sample.device
        ).view(batch_size)
        loss = (per_sample * weights).mean()

        # new priorities = |TD error|; device buffers take the tensor
        # directly (no D2H sync on the update path)
        abs_td = (q_taken - y).detach().abs().view(batch_size)
        if getattr(self.replay_buffer, "accepts_tensor_priorities", False):
            self.replay_buffer.update_priority(abs_td, index)
        else:
            self.replay_buffer.update_priority(
                abs_td.cpu().numpy().astype(np.float64), index
            )

        if self.visualize:
            self.visualize_model(loss, "qnet", self.visualize_dir)
        if update_value:
            self.qnet_optim.zero_grad(set_to_none=True)
            self._backward(loss)
            nn.utils.clip_grad_norm_(self.qnet.parameters(), self.grad_max)
            self.qnet_optim.step()
        if update_target:
            from .utils import hard_update, soft_update

            if self.update_rate is not None:
                soft_update(self.qnet_target, self.qnet, self.update_rate)
            else:
                self._update_counter += 1
                if self._update_counter % self.update_steps == 0:
                    hard_update(self.qnet_target, self.qnet)
        return float(loss.detach().item())
