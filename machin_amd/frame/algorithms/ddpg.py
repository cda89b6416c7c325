"""DDPG: deep deterministic policy gradient.

Parity target: reference ``machin/frame/algorithms/ddpg.py`` (DDPG
:30): same API — ``act``/``act_with_noise`` (4 noise modes),
``act_discrete(_with_noise)``, ``_criticize``, ``_act``, store/update,
``action_transform_function`` / ``action_concat_function`` /
``reward_function`` hooks, soft target updates.
"""
from typing import Any, Callable, Dict, List, Union

import numpy as np
import torch as t
import torch.nn as nn

from ...utils.conf import Config
from ..buffers.buffer import Buffer
from ..noise.action_space_noise import (
    add_clipped_normal_noise_to_action,
    add_normal_noise_to_action,
    add_ou_noise_to_action,
    add_uniform_noise_to_action,
)
from ..transition import Transition
from .base import TorchFramework
from .utils import (
    assert_and_get_valid_criterion,
    assert_and_get_valid_lr_scheduler,
    assert_and_get_valid_models,
    assert_and_get_valid_optimizer,
    hard_update,
    safe_call,
    safe_return,
    soft_update,
)


class DDPG(TorchFramework):
    """DDPG framework (continuous actions; discrete via argmax)."""

    _is_top = ["actor", "actor_target", "critic", "critic_target"]
    _is_restorable = ["actor_target", "critic_target"]

    def __init__(
        self,
        actor: nn.Module,
        actor_target: nn.Module,
        critic: nn.Module,
        critic_target: nn.Module,
        optimizer: Callable,
        criterion: Callable,
        *_,
        lr_scheduler: Callable = None,
        lr_scheduler_args: tuple = None,
        lr_scheduler_kwargs: tuple = None,
        batch_size: int = 100,
        update_rate: Union[float, None] = 0.005,
        update_steps: Union[int, None] = None,
        actor_learning_rate: float = 0.0005,
        critic_learning_rate: float = 0.001,
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
        self.actor_target = actor_target
        self.critic = critic
        self.critic_target = critic_target

        if isinstance(optimizer, (tuple, list)):
            actor_optim_cls, critic_optim_cls = optimizer
        else:
            actor_optim_cls = critic_optim_cls = optimizer
        self.actor_optim = actor_optim_cls(
            self.actor.parameters(), lr=actor_learning_rate
        )
        self.critic_optim = critic_optim_cls(
            self.critic.parameters(), lr=critic_learning_rate
        )
        # replay_device="cuda:*" routes replay to the HBM-resident
        # flat rings (machin_amd/frame/buffers/device_buffer.py)
        from ..buffers import default_buffer

        self.replay_buffer = (
            default_buffer(replay_size, replay_device)
            if replay_buffer is None
            else replay_buffer
        )

        hard_update(self.actor_target, self.actor)
        hard_update(self.critic_target, self.critic)
        if update_rate is not None and update_steps is not None:
            raise ValueError(
                "Specify either update_rate or update_steps, not both."
            )

        self.actor_lr_sch = None
        self.critic_lr_sch = None
        if lr_scheduler is not None:
            args = lr_scheduler_args or ((), ())
            kwargs = lr_scheduler_kwargs or ({}, {})
            self.actor_lr_sch = lr_scheduler(
                self.actor_optim, *args[0], **kwargs[0]
            )
            self.critic_lr_sch = lr_scheduler(
                self.critic_optim, *args[1], **kwargs[1]
            )
        self.criterion = criterion() if isinstance(criterion, type) else criterion

    # ------------------------------------------------------------------
    @property
    def optimizers(self):
        return [self.actor_optim, self.critic_optim]

    @optimizers.setter
    def optimizers(self, optimizers):
        self.actor_optim, self.critic_optim = optimizers

    @property
    def lr_schedulers(self):
        out = []
        if self.actor_lr_sch is not None:
            out.append(self.actor_lr_sch)
        if self.critic_lr_sch is not None:
            out.append(self.critic_lr_sch)
        return out

    # ------------------------------------------------------------------
    # acting
    # ------------------------------------------------------------------
    def act(self, state: Dict[str, Any], use_target: bool = False, **__):
        """Deterministic action from the (target) actor."""
        net = self.actor_target if use_target else self.actor
        return safe_return(safe_call(net, state))

    def act_with_noise(
        self,
        state: Dict[str, Any],
        noise_param: Any = (0.0, 1.0),
        ratio: float = 1.0,
        mode: str = "uniform",
        use_target: bool = False,
        **__,
    ):
        """Noisy action; ``mode`` in {uniform, normal, clipped_normal,
        ou}."""
        action = self.act(state, use_target)
        if mode == "uniform":
            return add_uniform_noise_to_action(action, noise_param, ratio)
        if mode == "normal":
            return add_normal_noise_to_action(action, noise_param, ratio)
        if mode == "clipped_normal":
            return add_clipped_normal_noise_to_action(
                action, noise_param, ratio
            )
        if mode == "ou":
            return add_ou_noise_to_action(action, noise_param, ratio)
        raise ValueError(f"Unknown noise type {mode!r}")

    def act_discrete(self, state: Dict[str, Any], use_target: bool = False,
                     **__):
        """Discrete action = argmax over actor output (probabilities);
        returns (action [B,1], probs)."""
        action = self.act(state, use_target)
        batch = action.shape[0]
        result = t.argmax(action, dim=1).view(batch, 1)
        return result, action

    def act_discrete_with_noise(
        self, state: Dict[str, Any], use_target: bool = False, **__
    ):
        """Sample a discrete action from actor-output probabilities."""
        action = self.act(state, use_target)
        batch = action.shape[0]
        dist = t.distributions.Categorical(probs=action)
        result = dist.sample([1]).view(batch, 1)
        return result, action

    def _criticize(
        self,
        state: Dict[str, Any],
        action: Dict[str, Any],
        use_target: bool = False,
        **__,
    ):
        net = self.critic_target if use_target else self.critic
        return safe_return(safe_call(net, state, action))

    # ------------------------------------------------------------------
    # storing
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
    # updating
    # ------------------------------------------------------------------
    def update(
        self,
        update_value=True,
        update_policy=True,
        update_target=True,
        concatenate_samples=True,
        **__,
    ):
        """One actor-critic update. Returns (-policy_loss_mean,
        value_loss)."""
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

        # critic target: y = r + discount*(1-terminal)*Q'(s', pi'(s'))
        with t.no_grad():
            next_action = self.action_transform_function(
                self.act(next_state, True), next_state, others
            )
            next_value = self._criticize(next_state, next_action, True)
            device = next_value.device
            reward = reward.to(device).float().view(batch_size, 1)
            terminal = terminal.to(device).float().view(batch_size, 1)
            y = self.reward_function(
                reward, self.discount, next_value.view(batch_size, 1), terminal,
                others,
            )

        cur_value = self._criticize(state, action)
        value_loss = self.criterion(cur_value, y.to(cur_value.dtype))
        if self.visualize:
            self.visualize_model(value_loss, "critic", self.visualize_dir)
        if update_value:
            self.critic_optim.zero_grad(set_to_none=True)
            self._backward(value_loss)
            nn.utils.clip_grad_norm_(self.critic.parameters(), self.grad_max)
            self.critic_optim.step()

        # actor: maximize Q(s, pi(s))
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

    def update_lr_scheduler(self):
        if self.actor_lr_sch is not None:
            self.actor_lr_sch.step()
        if self.critic_lr_sch is not None:
            self.critic_lr_sch.step()

    def load(self, model_dir, network_map=None, version=-1):
        super().load(model_dir, network_map, version)
        with t.no_grad():
            hard_update(self.actor, self.actor_target)
            hard_update(self.critic, self.critic_target)

    # ------------------------------------------------------------------
    # hooks
    # ------------------------------------------------------------------
    @staticmethod
    def action_transform_function(raw_output_action: t.Tensor, *_):
        """Map raw actor output to the critic's action-dict input."""
        return {"action": raw_output_action}

    @staticmethod
    def reward_function(reward, discount, next_value, terminal, _):
        return reward + discount * (1.0 - terminal) * next_value

    # ------------------------------------------------------------------
    @classmethod
    def generate_config(cls, config: Union[Dict[str, Any], Config]):
        default = {
            "frame": cls.__name__,
            "models": ["Actor", "Actor", "Critic", "Critic"],
            "model_args": ((), (), (), ()),
            "model_kwargs": ({}, {}, {}, {}),
            "optimizer": "Adam",
            "criterion": "MSELoss",
            "criterion_args": (),
            "criterion_kwargs": {},
            "lr_scheduler": None,
            "lr_scheduler_args": None,
            "lr_scheduler_kwargs": None,
            "batch_size": 100,
            "update_rate": 0.005,
            "update_steps": None,
            "actor_learning_rate": 0.0005,
            "critic_learning_rate": 0.001,
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
    def init_from_config(
        cls, config: Union[Dict[str, Any], Config], model_device="cpu"
    ):
        data = config.data if isinstance(config, Config) else dict(config)
        fc = data["frame_config"]
        model_cls = assert_and_get_valid_models(fc["models"])
        model_args = fc.get("model_args", ((), (), (), ()))
        model_kwargs = fc.get("model_kwargs", ({}, {}, {}, {}))
        models = [
            m(*args, **kwargs).to(model_device)
            for m, args, kwargs in zip(model_cls, model_args, model_kwargs)
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
                    "frame",
                    "models",
                    "model_args",
                    "model_kwargs",
                    "optimizer",
                    "criterion",
                    "criterion_args",
                    "criterion_kwargs",
                    "lr_scheduler",
                )
            },
        )
