"""GAIL: generative adversarial imitation learning.

Parity target: reference ``machin/frame/algorithms/gail.py``:
``ExpertTransition(state, action)`` records (:21-57), discriminator
BCE real-vs-generated (:273-306), reward shaping
``-log(1 - D(s,a))`` feeding an inner PPO or TRPO instance
(:308-313).
"""
from typing import Dict, List, Union

import torch as t
import torch.nn as nn

from ...utils.conf import Config
from ..buffers.buffer import Buffer
from ..transition import TransitionBase
from .base import TorchFramework
from .ppo import PPO
from .trpo import TRPO
from .utils import safe_call, safe_return


class ExpertTransition(TransitionBase):
    """A (state, action) expert demonstration record (batch size 1)."""

    def __init__(self, state: Dict[str, t.Tensor], action: Dict[str, t.Tensor]):
        super().__init__(
            major_attr=["state", "action"],
            sub_attr=[],
            custom_attr=[],
            major_data=[state, action],
            sub_data=[],
            custom_data=[],
        )


class GAIL(TorchFramework):
    _is_top = ["discriminator"]
    _is_restorable = ["discriminator"]

    def __init__(
        self,
        discriminator: nn.Module,
        constrained_policy_optimization: Union[PPO, TRPO],
        optimizer,
        *_,
        discriminator_learning_rate: float = 0.001,
        gradient_max: float = 1e9,
        expert_replay_size: int = 500000,
        expert_replay_device: Union[str, t.device] = "cpu",
        expert_replay_buffer: Buffer = None,
        visualize: bool = False,
        visualize_dir: str = "",
        **__,
    ):
        super().__init__()
        if not isinstance(constrained_policy_optimization, (PPO, TRPO)):
            raise ValueError(
                "constrained_policy_optimization must be a PPO or TRPO "
                "instance."
            )
        self.discriminator = discriminator
        self.cpo = constrained_policy_optimization
        self.grad_max = gradient_max
        self.visualize = visualize
        self.visualize_dir = visualize_dir
        self.discriminator_optim = optimizer(
            discriminator.parameters(), lr=discriminator_learning_rate
        )
        self.expert_replay_buffer = (
            Buffer(expert_replay_size, expert_replay_device)
            if expert_replay_buffer is None
            else expert_replay_buffer
        )
        self.bce = nn.BCELoss()
        # register inner frameworks' models as restorable via attribute
        self._is_restorable = ["discriminator"] + [
            "cpo_" + n for n in type(self.cpo)._is_restorable
        ]
        for n in type(self.cpo)._is_restorable:
            setattr(self, "cpo_" + n, getattr(self.cpo, n))

    @property
    def optimizers(self):
        return [self.discriminator_optim] + list(self.cpo.optimizers)

    @optimizers.setter
    def optimizers(self, optimizers):
        self.discriminator_optim = optimizers[0]
        self.cpo.optimizers = optimizers[1:]

    @property
    def lr_schedulers(self):
        return self.cpo.lr_schedulers

    # ------------------------------------------------------------------
    def act(self, state: Dict[str, t.Tensor], **__):
        return self.cpo.act(state)

    def _discriminate(self, state, action, **__):
        return safe_return(safe_call(self.discriminator, state, action))

    # ------------------------------------------------------------------
    def store_expert_episode(self, episode: List[Union[ExpertTransition, Dict]]):
        episode = [
            ExpertTransition(**tr) if isinstance(tr, dict) else tr
            for tr in episode
        ]
        self.expert_replay_buffer.store_episode(
            episode, required_attrs=("state", "action")
        )

    def store_episode(self, episode: List[Dict]):
        """Replace env rewards with discriminator rewards
        ``-log(1 - D(s,a))``, then hand the episode to the inner
        PPO/TRPO."""
        shaped = []
        for tr in episode:
            d = (
                {k: v for k, v in tr.items()}
                if isinstance(tr, dict)
                else {k: getattr(tr, k) for k in tr.keys()}
            )
            with t.no_grad():
                prob = self._discriminate(d["state"], d["action"]).view(-1)[0]
            d["reward"] = float(-t.log(1.0 - prob.clamp(0.0, 0.999)).item())
            shaped.append(d)
        self.cpo.store_episode(shaped)

    # ------------------------------------------------------------------
    def update(self, update_discriminator=True, update_policy=True, **__):
        """Train the discriminator on expert-vs-generated pairs, then
        run the inner policy update. Returns (policy metrics...,
        discriminator_loss)."""
        d_loss_val = 0.0
        if update_discriminator:
            # generated pairs: everything currently in the cpo buffer
            bs_g, gen = self.cpo.replay_buffer.sample_batch(
                -1, sample_method="all", concatenate=True,
                sample_attrs=["state", "action"],
            )
            bs_e, exp = self.expert_replay_buffer.sample_batch(
                bs_g if bs_g > 0 else 100,
                sample_method="random",
                concatenate=True,
                sample_attrs=["state", "action"],
            )
            if bs_g > 0 and bs_e > 0:
                self.discriminator.train()
                gen_prob = self._discriminate(*gen).view(bs_g, 1)
                exp_prob = self._discriminate(*exp).view(bs_e, 1)
                d_loss = self.bce(
                    exp_prob, t.ones_like(exp_prob)
                ) + self.bce(gen_prob, t.zeros_like(gen_prob))
                if self.visualize:
                    self.visualize_model(
                        d_loss, "discriminator", self.visualize_dir
                    )
                self.discriminator_optim.zero_grad(set_to_none=True)
                self._backward(d_loss)
                nn.utils.clip_grad_norm_(
                    self.discriminator.parameters(), self.grad_max
                )
                self.discriminator_optim.step()
                d_loss_val = float(d_loss.detach().item())

        result = self.cpo.update(**__) if update_policy else ()
        return (*result, d_loss_val)

    def update_lr_scheduler(self):
        self.cpo.update_lr_scheduler()

    # ------------------------------------------------------------------
    @classmethod
    def generate_config(cls, config):
        config = PPO.generate_config(config)
        config["frame"] = "GAIL"
        fc = config["frame_config"]
        fc["frame"] = "GAIL"
        fc.setdefault("models", ["Actor", "Critic", "Discriminator"])
        fc.setdefault("discriminator_learning_rate", 0.001)
        fc.setdefault("expert_replay_size", 500000)
        fc.setdefault("expert_replay_device", "cpu")
        return config

    @classmethod
    def init_from_config(cls, config, model_device="cpu"):
        data = config.data if isinstance(config, Config) else dict(config)
        fc = dict(data["frame_config"])
        models = list(fc["models"])
        disc_name = models.pop()  # last model is the discriminator
        from .utils import (
            assert_and_get_valid_models,
            assert_and_get_valid_optimizer,
        )

        inner_conf = {
            "frame_config": {**fc, "models": models,
                             "model_args": fc.get("model_args", ((),) * 3)[:2],
                             "model_kwargs": fc.get("model_kwargs", ({},) * 3)[:2]}
        }
        cpo = PPO.init_from_config(PPO.generate_config(inner_conf), model_device)
        disc_cls = assert_and_get_valid_models([disc_name])[0]
        disc_args = fc.get("model_args", ((), (), ()))[-1]
        disc_kwargs = fc.get("model_kwargs", ({}, {}, {}))[-1]
        discriminator = disc_cls(*disc_args, **disc_kwargs).to(model_device)
        optimizer = assert_and_get_valid_optimizer(fc["optimizer"])
        return cls(
            discriminator,
            cpo,
            optimizer,
            discriminator_learning_rate=fc.get(
                "discriminator_learning_rate", 0.001
            ),
            expert_replay_size=fc.get("expert_replay_size", 500000),
            expert_replay_device=fc.get("expert_replay_device", "cpu"),
        )
