"""RAINBOW: distributional C51 + prioritized replay + n-step returns.

Parity target: reference ``machin/frame/algorithms/rainbow.py``
(:179-301): n-step reward folding at ``store_episode`` time
(machin_amd.ops.nstep_returns), categorical projection of
``r + gamma^n * z`` onto the fixed support (gfx950 kernel
machin_amd.ops.categorical_projection on GPU), cross-entropy loss
weighted by IS weights, abs-TD-style priority write-back.

Model contract: ``qnet(state) -> [batch, action_num, atom_num]``
distribution (softmaxed over atoms).
"""
from typing import Any, Dict, List, Union

import numpy as np
import torch as t
import torch.nn as nn

from ... import ops
from ..buffers.prioritized_buffer import PrioritizedBuffer
from ..transition import Transition
from .dqn_per import DQNPer
from .utils import hard_update, safe_call, safe_return, soft_update


class RAINBOW(DQNPer):
    def __init__(
        self,
        qnet,
        qnet_target,
        optimizer,
        value_min: float,
        value_max: float,
        *_,
        criterion=None,
        reward_future_steps: int = 3,
        replay_size: int = 500000,
        replay_device: Union[str, t.device] = "cpu",
        replay_buffer=None,
        **kwargs,
    ):
        super().__init__(
            qnet,
            qnet_target,
            optimizer,
            criterion or nn.MSELoss(reduction="none"),
            replay_size=replay_size,
            replay_device=replay_device,
            replay_buffer=replay_buffer,
            **kwargs,
        )
        self.v_min = float(value_min)
        self.v_max = float(value_max)
        self.reward_future_steps = reward_future_steps

    # ------------------------------------------------------------------
    def _criticize(self, state: Dict[str, Any], use_target: bool = False,
                   **__):
        """Returns the distribution [B, A, N]."""
        net = self.qnet_target if use_target else self.qnet
        return safe_return(safe_call(net, state))

    def _q_values(self, dist: t.Tensor) -> t.Tensor:
        B, A, N = dist.shape
        support = t.linspace(
            self.v_min, self.v_max, N, device=dist.device
        )
        return (dist * support.view(1, 1, N)).sum(dim=2)

    def act_discrete(self, state, use_target=False, **__):
        q = self._q_values(self._criticize(state, use_target))
        return t.argmax(q, dim=1).view(-1, 1)

    def act_discrete_with_noise(self, state, use_target=False,
                                decay_epsilon=True, **__):
        q = self._q_values(self._criticize(state, use_target))
        batch, n_actions = q.shape
        if t.rand(1).item() < self.epsilon:
            result = t.randint(0, n_actions, (batch, 1))
        else:
            result = t.argmax(q, dim=1).view(-1, 1)
        if decay_epsilon:
            self.epsilon *= self.epsilon_decay
        return result

    # ------------------------------------------------------------------
    def store_episode(self, episode: List[Union[Transition, Dict]]):
        """Fold the next ``reward_future_steps`` rewards into each
        transition before storing (n-step returns)."""
        episode = [
            Transition(**tr) if isinstance(tr, dict) else tr for tr in episode
        ]
        n = self.reward_future_steps
        if n > 1:
            T = len(episode)
            rewards = t.tensor(
                [float(tr.reward) if not t.is_tensor(tr.reward)
                 else float(tr.reward.reshape(-1)[0]) for tr in episode]
            )
            terminals = t.tensor(
                [float(tr.terminal) if not t.is_tensor(tr.terminal)
                 else float(tr.terminal.reshape(-1)[0]) for tr in episode]
            )
            nstep = ops.nstep_returns(rewards, terminals, self.discount, n)
            # single host copy for the whole episode (no per-element
            # .item() synchronization)
            nstep_list = nstep.tolist()
            term_list = terminals.tolist()
            new_episode = []
            for i, tr in enumerate(episode):
                d = {k: getattr(tr, k) for k in tr.keys()}
                d["reward"] = float(nstep_list[i])
                # bootstrap state is s_{i+n} (or the episode end)
                j = min(i + n - 1, T - 1)
                d["next_state"] = {
                    k: v for k, v in episode[j].next_state.items()
                }
                d["terminal"] = bool(max(term_list[i : j + 1]) > 0.5)
                new_episode.append(Transition(**d))
            episode = new_episode
        self.replay_buffer.store_episode(
            episode,
            required_attrs=("state", "action", "next_state", "reward", "terminal"),
        )

    # ------------------------------------------------------------------
    def update(self, update_value=True, update_target=True,
               concatenate_samples=True, **__):
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

        dist = self._criticize(state)  # [B, A, N]
        B, A, N = dist.shape
        device = dist.device
        action_index = action["action"].to(device=device, dtype=t.long).view(B)

        with t.no_grad():
            # double-DQN action selection with the online net
            online_next = self._criticize(next_state)
            next_q = self._q_values(online_next)
            best = next_q.argmax(dim=1)  # [B]
            target_next = self._criticize(next_state, use_target=True)
            next_dist = target_next[t.arange(B, device=device), best]  # [B,N]
            reward = reward.to(device).float().view(B)
            terminal = terminal.to(device).float().view(B)
            gamma_n = self.discount ** self.reward_future_steps
            proj = ops.categorical_projection(
                next_dist, reward, terminal, gamma_n, self.v_min, self.v_max
            )

        pred = dist[t.arange(B, device=device), action_index]  # [B,N]
        log_pred = t.log(pred.clamp_min(1e-8))
        per_sample = -(proj * log_pred).sum(dim=1)  # cross entropy
        weights = t.as_tensor(
            is_weight, dtype=per_sample.dtype, device=device
        ).view(B)
        loss = (per_sample * weights).mean()

        if getattr(self.replay_buffer, "accepts_tensor_priorities", False):
            self.replay_buffer.update_priority(per_sample.detach(), index)
        else:
            self.replay_buffer.update_priority(
                per_sample.detach().cpu().numpy().astype(np.float64), index
            )

        if self.visualize:
            self.visualize_model(loss, "qnet", self.visualize_dir)
        if update_value:
            self.qnet_optim.zero_grad(set_to_none=True)
            self._backward(loss)
            nn.utils.clip_grad_norm_(self.qnet.parameters(), self.grad_max)
            self.qnet_optim.step()
        if update_target:
            if self.update_rate is not None:
                soft_update(self.qnet_target, self.qnet, self.update_rate)
            else:
                self._update_counter += 1
                if self._update_counter % self.update_steps == 0:
                    hard_update(self.qnet_target, self.qnet)
        return float(loss.detach().item())

    # ------------------------------------------------------------------
    @classmethod
    def generate_config(cls, config):
        from .dqn import DQN

        config = DQN.generate_config(config)
        config["frame"] = "RAINBOW"
        fc = config["frame_config"]
        fc["frame"] = "RAINBOW"
        fc.pop("mode", None)
        fc.setdefault("value_min", -10.0)
        fc.setdefault("value_max", 10.0)
        fc.setdefault("reward_future_steps", 3)
        return config

    @classmethod
    def init_from_config(cls, config, model_device="cpu"):
        from ...utils.conf import Config
        from .utils import (
            assert_and_get_valid_criterion,
            assert_and_get_valid_models,
            assert_and_get_valid_optimizer,
        )

        data = config.data if isinstance(config, Config) else dict(config)
        fc = data["frame_config"]
        model_cls = assert_and_get_valid_models(fc["models"])
        models = [
            m(*args, **kwargs).to(model_device)
            for m, args, kwargs in zip(
                model_cls, fc.get("model_args", ((), ())),
                fc.get("model_kwargs", ({}, {})),
            )
        ]
        optimizer = assert_and_get_valid_optimizer(fc["optimizer"])
        criterion = assert_and_get_valid_criterion(fc["criterion"])(
            *fc.get("criterion_args", ()), **fc.get("criterion_kwargs", {})
        )
        return cls(
            models[0], models[1], optimizer,
            fc["value_min"], fc["value_max"],
            criterion=criterion,
            **{
                k: v
                for k, v in fc.items()
                if k
                not in (
                    "frame", "models", "model_args", "model_kwargs",
                    "optimizer", "criterion", "criterion_args",
                    "criterion_kwargs", "lr_scheduler", "value_min",
                    "value_max", "mode",
                )
            },
        )
