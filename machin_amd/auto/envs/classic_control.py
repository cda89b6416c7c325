"""Auto-layer plugin for the built-in classic-control environments.

Parity target: reference ``machin/auto/envs/openai_gym.py``: episode
datasets with per-algorithm act dispatch (:83-140, :192-246) and the
``launch`` entry (:295-343). gym is absent in the ROCm image, so the
built-in CartPole/Pendulum dynamics are used; the act-type dispatch
table covers every algorithm family.
"""
from typing import Any, Dict

import torch as t

from ...env.envs.classic_control import make
from ...frame.algorithms import (
    A2C,
    A3C,
    ARS,
    DDPG,
    DDPGPer,
    DQN,
    DQNPer,
    GAIL,
    HDDPG,
    IMPALA,
    PPO,
    RAINBOW,
    SAC,
    TD3,
)
from ..dataset import DatasetResult, RLDataset

_DISCRETE_DETERMINISTIC = (DQN, DQNPer, RAINBOW)
_STOCHASTIC = (A2C, A3C, PPO, IMPALA, GAIL)
_CONTINUOUS_DETERMINISTIC = (DDPG, DDPGPer, HDDPG, TD3)


def _env_name(config) -> str:
    return config.get("train_env_config", {}).get(
        "env_name", config.get("env_name", "CartPole-v1")
    )


class RLGymDiscActDataset(RLDataset):
    """One episode per item for discrete-action algorithms."""

    def __init__(self, frame, env, act_kwargs: Dict[str, Any] = None):
        super().__init__()
        self.frame = frame
        self.env = env
        self.act_kwargs = act_kwargs or {}

    def __next__(self) -> DatasetResult:
        result = DatasetResult()
        obs = t.tensor(self.env.reset(), dtype=t.float32).view(1, -1)
        total_reward = 0.0
        transitions = []
        done = False
        while not done:
            with t.no_grad():
                if isinstance(self.frame, _DISCRETE_DETERMINISTIC):
                    action = self.frame.act_discrete_with_noise(
                        {"state": obs}, **self.act_kwargs
                    )
                elif isinstance(self.frame, _STOCHASTIC):
                    action = self.frame.act({"state": obs})[0]
                else:
                    raise RuntimeError(
                        f"Unsupported frame {type(self.frame).__name__} "
                        f"for discrete environments."
                    )
            obs_next, reward, done, _ = self.env.step(int(action.item()))
            obs_next = t.tensor(obs_next, dtype=t.float32).view(1, -1)
            total_reward += reward
            tr = {
                "state": {"state": obs},
                "action": {"action": action.view(1, 1)},
                "next_state": {"state": obs_next},
                "reward": float(reward),
                "terminal": done
                and self.env.steps < self.env.max_episode_steps,
            }
            if isinstance(self.frame, IMPALA):
                with t.no_grad():
                    lp = self.frame._eval_act(
                        tr["state"], tr["action"]
                    )[1]
                tr["action_log_prob"] = float(lp.item())
            transitions.append(tr)
            obs = obs_next
        result.add_observation(transitions)
        result.add_log({"total_reward": total_reward})
        return result


class RLGymContActDataset(RLDataset):
    """One episode per item for continuous-action algorithms."""

    def __init__(self, frame, env, act_kwargs: Dict[str, Any] = None):
        super().__init__()
        self.frame = frame
        self.env = env
        self.act_kwargs = act_kwargs or {"noise_param": (0.0, 0.3),
                                         "mode": "normal"}

    def __next__(self) -> DatasetResult:
        result = DatasetResult()
        obs = t.tensor(self.env.reset(), dtype=t.float32).view(1, -1)
        total_reward = 0.0
        transitions = []
        done = False
        lo = float(self.env.action_space.low[0])
        hi = float(self.env.action_space.high[0])
        while not done:
            with t.no_grad():
                if isinstance(self.frame, _CONTINUOUS_DETERMINISTIC):
                    action = self.frame.act_with_noise(
                        {"state": obs}, **self.act_kwargs
                    ).clamp(lo, hi)
                elif isinstance(self.frame, SAC):
                    action = self.frame.act({"state": obs})[0].clamp(lo, hi)
                else:
                    raise RuntimeError(
                        f"Unsupported frame {type(self.frame).__name__} "
                        f"for continuous environments."
                    )
            obs_next, reward, done, _ = self.env.step(
                action.view(-1).numpy()
            )
            obs_next = t.tensor(obs_next, dtype=t.float32).view(1, -1)
            total_reward += reward
            transitions.append(
                {
                    "state": {"state": obs},
                    "action": {"action": action.view(1, -1)},
                    "next_state": {"state": obs_next},
                    "reward": float(reward),
                    "terminal": done
                    and self.env.steps < self.env.max_episode_steps,
                }
            )
            obs = obs_next
        result.add_observation(transitions)
        result.add_log({"total_reward": total_reward})
        return result


def generate_env_config(environment: str, config=None):
    """Attach environment settings to a config."""
    from ...utils.conf import Config

    config = config if config is not None else Config()
    config["env"] = "classic_control"
    config["train_env_config"] = {"env_name": environment}
    config["test_env_config"] = {"env_name": environment}
    return config


def make_dataset(frame, config) -> RLDataset:
    env = make(_env_name(config))
    base = _env_name(config).split("-")[0].lower()
    if base in ("cartpole", "pixelcatch"):
        return RLGymDiscActDataset(frame, env)
    return RLGymContActDataset(frame, env)


def launch(config, checkpoint_callback=None, early_stopping=None):
    """Build the frame + dataset and run the Launcher fit loop."""
    from ..launcher import Launcher

    launcher = Launcher(config, dataset_factory=make_dataset)
    return launcher.fit()
