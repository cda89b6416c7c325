"""SAC: soft actor-critic with automatic entropy tuning.

Parity target: reference ``machin/frame/algorithms/sac.py`` (:130-384):
twin critics with targets (no actor target), actor contract
``forward(state) -> (action, log_prob, ...)``, soft value target
``min(Q1', Q2') - alpha * log_prob``, optional learnable
``entropy_alpha`` against ``target_entropy``.
"""
from typing import Any, Callable, Dict, List, Union

import numpy as np
import torch as t
import torch.nn as nn

from ...utils.conf import Config
from ..buffers.buffer import Buffer
from ..transition import Transition
from .base import TorchFramework
from .utils import (
    assert_and_get_valid_criterion,
    assert_and_get_valid_lr_scheduler,
    assert_and_get_valid_models,
    assert_and_get_valid_optimizer,
    hard_update,
    safe_call,
    soft_update,
)


class SAC(TorchFramework):
    _is_top = ["actor", "critic", "critic_target", "critic2", "critic2_target"]
    _is_restorable = ["actor", "critic_target", "critic2_target"]

    def __init__(
        self,
        actor: nn.Module,
        critic: nn.Module,
        critic_target: nn.Module,
        critic2: nn.Module,
        critic2_target: nn.Module,
        optimizer: Callable,
        criterion: Callable,
        *_,
        lr_scheduler: Callable = None,
        lr_scheduler_args: tuple = None,
        lr_scheduler_kwargs: tuple = None,
        target_entropy: float = None,
        initial_entropy_alpha: float = 1.0,
        batch_size: int = 100,
        update_rate: Union[float, None] = 0.005,
        update_steps: Union[int, None] = None,
        actor_learning_rate: float = 0.0005,
        critic_learning_rate: float = 0.001,
        alpha_learning_rate: float = 0.001,
        discount: float = 0.99,
        gradient_max: float = np.inf,
        replay_size: int = 500000,
        replay_device: Union[str, t.device] = "cpu",
        replay_buffer: Buffer = None,
        visualize: bool = False,
        visualize_dir: str = "",
        **__,
    ):
        super().__init__()
        self.batch_size = batch_size
        self.update_rate = update_rate
        self.update_steps = update_steps
        self.discount = discount
        self.grad_max = gradient_max
        self.visualize = visualize
        self.visualize_dir = visualize_dir
        self._update_counter = 0

        self.actor = actor
        self.critic = critic
        self.critic_target = critic_target
        self.critic2 = critic2
        self.critic2_target = critic2_target
        self.target_entropy = target_entropy
        self.entropy_alpha = t.tensor(
            [float(initial_entropy_alpha)], requires_grad=True
        )

        if isinstance(optimizer, (tuple, list)):
            a_opt, c_opt = optimizer[0], optimizer[1]
        else:
            a_opt = c_opt = optimizer
        self.actor_optim = a_opt(self.actor.parameters(), lr=actor_learning_rate)
        self.critic_optim = c_opt(
            self.critic.parameters(), lr=critic_learning_rate
        )
        self.critic2_optim = c_opt(
            self.critic2.parameters(), lr=critic_learning_rate
        )
        self.alpha_optim = t.optim.Adam(
            [self.entropy_alpha], lr=alpha_learning_rate
        )
        # replay_device="cuda:*" routes replay to the HBM-resident
        # flat rings (machin_amd/frame/buffers/device_buffer.py)
        from ..buffers import default_buffer

        self.replay_buffer = (
            default_buffer(replay_size, replay_device)
            if replay_buffer is None
            else replay_buffer
        )
        hard_update(self.critic_target, self.critic)
        hard_update(self.critic2_target, self.critic2)

        self.actor_lr_sch = None
        self.critic_lr_sch = None
        if lr_scheduler is not None:
            args = lr_scheduler_args or ((), ())
            kwargs = lr_scheduler_kwargs or ({}, {})
            self.actor_lr_sch = lr_scheduler(self.actor_optim, *args[0], **kwargs[0])
            self.critic_lr_sch = lr_scheduler(self.critic_optim, *args[1], **kwargs[1])
        self.criterion = criterion() if isinstance(criterion, type) else criterion

    @property
    def optimizers(self):
        return [self.actor_optim, self.critic_optim, self.critic2_optim]

    @optimizers.setter
    def optimizers(self, optimizers):
        self.actor_optim, self.critic_optim, self.critic2_optim = optimizers

    @property
    def lr_schedulers(self):
        out = []
        if self.actor_lr_sch is not None:
            out.append(self.actor_lr_sch)
        if self.critic_lr_sch is not None:
            out.append(self.critic_lr_sch)
        return out

    # ------------------------------------------------------------------
    def act(self, state: Dict[str, Any], **__):
        """Returns the actor's full output tuple
        ``(action, log_prob, ...)``."""
        result = safe_call(self.actor, state)
        if not isinstance(result, tuple):
            raise ValueError(
                "SAC actor must return a tuple (action, log_prob, ...)."
            )
        return result

    def _criticize(self, state, action, use_target=False, **__):
        net = self.critic_target if use_target else self.critic
        result = safe_call(net, state, action)
        return result[0] if isinstance(result, tuple) else result

    def _criticize2(self, state, action, use_target=False, **__):
        net = self.critic2_target if use_target else self.critic2
        result = safe_call(net, state, action)
        return result[0] if isinstance(result, tuple) else result

    # ------------------------------------------------------------------
    def store_transition(self, transition: Union[Transition, Dict]):
        self.replay_buffer.store_episode(
            [transition],
            required_attrs=("state", "action", "next_state", "reward", "terminal"),
        )

    def store_episode(self, episode: List[Union[Transition, Dict]]):
        self.replay_buffer.store_episode(
            episode,
            required_attrs=("state", "action", "next_state", "reward", "terminal"),
        )

    # ------------------------------------------------------------------
    def update(
        self,
        update_value=True,
        update_policy=True,
        update_target=True,
        update_entropy_alpha=True,
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
        alpha = self.entropy_alpha.detach()

        # critic target
        with t.no_grad():
            next_result = self.act(next_state)
            next_action = self.action_transform_function(
                next_result[0], next_state, others
            )
            next_log_prob = next_result[1]
            q1 = self._criticize(next_state, next_action, True)
            q2 = self._criticize2(next_state, next_action, True)
            device = q1.device
            soft_q = t.min(q1, q2).view(batch_size, 1) - alpha.to(
                device
            ) * next_log_prob.view(batch_size, 1)
            reward = reward.to(device).float().view(batch_size, 1)
            terminal = terminal.to(device).float().view(batch_size, 1)
            y = self.reward_function(
                reward, self.discount, soft_q, terminal, others
            )

        cur_v1 = self._criticize(state, action)
        cur_v2 = self._criticize2(state, action)
        value_loss1 = self.criterion(cur_v1, y.to(cur_v1.dtype))
        value_loss2 = self.criterion(cur_v2, y.to(cur_v2.dtype))
        if self.visualize:
            self.visualize_model(value_loss1, "critic", self.visualize_dir)
        if update_value:
            self.critic_optim.zero_grad(set_to_none=True)
            self._backward(value_loss1)
            nn.utils.clip_grad_norm_(self.critic.parameters(), self.grad_max)
            self.critic_optim.step()
            self.critic2_optim.zero_grad(set_to_none=True)
            self._backward(value_loss2)
            nn.utils.clip_grad_norm_(self.critic2.parameters(), self.grad_max)
            self.critic2_optim.step()

        # actor
        result = self.act(state)
        cur_action = self.action_transform_function(result[0], state, others)
        log_prob = result[1]
        q1 = self._criticize(state, cur_action)
        q2 = self._criticize2(state, cur_action)
        q_min = t.min(q1, q2).view(batch_size, 1)
        act_policy_loss = (
            alpha.to(q_min.device) * log_prob.view(batch_size, 1) - q_min
        ).mean()
        if self.visualize:
            self.visualize_model(act_policy_loss, "actor", self.visualize_dir)
        if update_policy:
            self.actor_optim.zero_grad(set_to_none=True)
            self._backward(act_policy_loss)
            nn.utils.clip_grad_norm_(self.actor.parameters(), self.grad_max)
            self.actor_optim.step()

        if update_target:
            if self.update_rate is not None:
                soft_update(self.critic_target, self.critic, self.update_rate)
                soft_update(
                    self.critic2_target, self.critic2, self.update_rate
                )
            else:
                self._update_counter += 1
                if self._update_counter % self.update_steps == 0:
                    hard_update(self.critic_target, self.critic)
                    hard_update(self.critic2_target, self.critic2)

        if update_entropy_alpha and self.target_entropy is not None:
            alpha_loss = -(
                t.log(self.entropy_alpha.clamp(1e-8, 1e8))
                * (log_prob.detach().cpu() + self.target_entropy).mean()
            )
            self.alpha_optim.zero_grad(set_to_none=True)
            alpha_loss.backward()
            self.alpha_optim.step()
            with t.no_grad():
                self.entropy_alpha.clamp_(1e-8, 1e8)

        return (
            -float(act_policy_loss.detach().item()),
            float(((value_loss1 + value_loss2) / 2).detach().item()),
        )

    def update_lr_scheduler(self):
        if self.actor_lr_sch is not None:
            self.actor_lr_sch.step()
        if self.critic_lr_sch is not None:
            self.critic_lr_sch.step()

    def load(self, model_dir, network_map=None, version=-1):
        super().load(model_dir, network_map, version)
        with t.no_grad():
            hard_update(self.critic, self.critic_target)
            hard_update(self.critic2, self.critic2_target)

    @staticmethod
    def action_transform_function(raw_output_action: t.Tensor, *_):
        return {"action": raw_output_action}

    @staticmethod
    def reward_function(reward, discount, next_value, terminal, _):
        return reward + discount * (1.0 - terminal) * next_value

    # ------------------------------------------------------------------
    @classmethod
    def generate_config(cls, config: Union[Dict[str, Any], Config]):
        default = {
            "frame": "SAC",
            "models": ["Actor", "Critic", "Critic", "Critic", "Critic"],
            "model_args": ((), (), (), (), ()),
            "model_kwargs": ({}, {}, {}, {}, {}),
            "optimizer": "Adam",
            "criterion": "MSELoss",
            "criterion_args": (),
            "criterion_kwargs": {},
            "lr_scheduler": None,
            "lr_scheduler_args": None,
            "lr_scheduler_kwargs": None,
            "target_entropy": None,
            "initial_entropy_alpha": 1.0,
            "batch_size": 100,
            "update_rate": 0.005,
            "update_steps": None,
            "actor_learning_rate": 0.0005,
            "critic_learning_rate": 0.001,
            "alpha_learning_rate": 0.001,
            "discount": 0.99,
            "gradient_max": 1e9,
            "replay_size": 500000,
            "replay_device": "cpu",
            "visualize": False,
            "visualize_dir": "",
        }
        config = config or {}
        data = config.data if isinstance(config, Config) else dict(config)
        frame_config = dict(default)
        frame_config.update(data.get("frame_config", {}))
        data["frame"] = frame_config["frame"]
        data["frame_config"] = frame_config
        return Config(**data)

    @classmethod
    def init_from_config(cls, config, model_device="cpu"):
        data = config.data if isinstance(config, Config) else dict(config)
        fc = data["frame_config"]
        model_cls = assert_and_get_valid_models(fc["models"])
        models = [
            m(*args, **kwargs).to(model_device)
            for m, args, kwargs in zip(
                model_cls, fc.get("model_args", ((),) * 5),
                fc.get("model_kwargs", ({},) * 5),
            )
        ]
        optimizer = assert_and_get_valid_optimizer(fc["optimizer"])
        criterion = assert_and_get_valid_criterion(fc["criterion"])(
            *fc.get("criterion_args", ()), **fc.get("criterion_kwargs", {})
        )
        lr_scheduler = assert_and_get_valid_lr_scheduler(fc.get("lr_scheduler"))
        return cls(
            *models,
            optimizer=optimizer,
            criterion=criterion,
            lr_scheduler=lr_scheduler,
            **{
                k: v
                for k, v in fc.items()
                if k
                not in (
                    "frame", "models", "model_args", "model_kwargs",
                    "optimizer", "criterion", "criterion_args",
                    "criterion_kwargs", "lr_scheduler",
                )
            },
        )
