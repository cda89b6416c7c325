"""Built-in classic-control environments.

The ROCm image has no gym/gymnasium, so machin_amd ships its own
CartPole and Pendulum dynamics (standard textbook equations of motion,
same observation/action/reward/termination contract as the OpenAI
implementations the reference's CI trains on — SURVEY.md §6 targets:
CartPole smoothed reward > 150, Pendulum > −400).

API is gym-classic: ``reset() -> obs``, ``step(a) -> (obs, reward,
done, info)``, ``seed``, ``render`` (no-op), ``close``; plus
``observation_space`` / ``action_space`` described by lightweight
``Space`` objects.
"""
import math
from typing import Optional

import numpy as np


class Space:
    """Minimal space descriptor (discrete or box)."""

    def __init__(self, shape=None, n=None, low=None, high=None, seed=None):
        self.shape = shape
        self.n = n
        self.low = low
        self.high = high
        self._rng = np.random.RandomState(seed)

    @property
    def discrete(self):
        return self.n is not None

    def sample(self):
        if self.discrete:
            return int(self._rng.randint(self.n))
        return self._rng.uniform(self.low, self.high, size=self.shape).astype(
            np.float32
        )

    def seed(self, seed):
        self._rng = np.random.RandomState(seed)


class CartPoleEnv:
    """CartPole-v1: balance a pole on a cart; +1 reward per step,
    episode ends on |x|>2.4, |theta|>12deg or 500 steps."""

    max_episode_steps = 500

    def __init__(self, seed: Optional[int] = None):
        self.gravity = 9.8
        self.masscart = 1.0
        self.masspole = 0.1
        self.total_mass = self.masspole + self.masscart
        self.length = 0.5
        self.polemass_length = self.masspole * self.length
        self.force_mag = 10.0
        self.tau = 0.02
        self.x_threshold = 2.4
        self.theta_threshold = 12 * 2 * math.pi / 360
        self.observation_space = Space(shape=(4,))
        self.action_space = Space(n=2)
        self._rng = np.random.RandomState(seed)
        self.state = None
        self.steps = 0

    def seed(self, seed=None):
        self._rng = np.random.RandomState(seed)
        self.action_space.seed(seed)
        return [seed]

    def reset(self):
        self.state = self._rng.uniform(-0.05, 0.05, size=(4,))
        self.steps = 0
        return self.state.astype(np.float32)

    def step(self, action):
        action = int(action)
        x, x_dot, theta, theta_dot = self.state
        force = self.force_mag if action == 1 else -self.force_mag
        costheta, sintheta = math.cos(theta), math.sin(theta)
        temp = (
            force + self.polemass_length * theta_dot ** 2 * sintheta
        ) / self.total_mass
        thetaacc = (self.gravity * sintheta - costheta * temp) / (
            self.length
            * (4.0 / 3.0 - self.masspole * costheta ** 2 / self.total_mass)
        )
        xacc = temp - self.polemass_length * thetaacc * costheta / self.total_mass
        x = x + self.tau * x_dot
        x_dot = x_dot + self.tau * xacc
        theta = theta + self.tau * theta_dot
        theta_dot = theta_dot + self.tau * thetaacc
        self.state = np.array([x, x_dot, theta, theta_dot])
        self.steps += 1
        done = (
            abs(x) > self.x_threshold
            or abs(theta) > self.theta_threshold
            or self.steps >= self.max_episode_steps
        )
        return self.state.astype(np.float32), 1.0, bool(done), {}

    def render(self, *_, **__):
        return None

    def close(self):
        pass


class PendulumEnv:
    """Pendulum-v1: swing up an underactuated pendulum; continuous
    torque in [-2, 2]; reward = -(theta^2 + 0.1*thdot^2 + 0.001*u^2);
    200-step episodes, never terminates early."""

    max_episode_steps = 200

    def __init__(self, seed: Optional[int] = None):
        self.max_speed = 8.0
        self.max_torque = 2.0
        self.dt = 0.05
        self.g = 10.0
        self.m = 1.0
        self.l = 1.0
        self.observation_space = Space(
            shape=(3,), low=np.array([-1, -1, -8.0]), high=np.array([1, 1, 8.0])
        )
        self.action_space = Space(
            shape=(1,),
            low=np.array([-self.max_torque]),
            high=np.array([self.max_torque]),
        )
        self._rng = np.random.RandomState(seed)
        self.state = None
        self.steps = 0

    def seed(self, seed=None):
        self._rng = np.random.RandomState(seed)
        self.action_space.seed(seed)
        return [seed]

    def reset(self):
        high = np.array([np.pi, 1.0])
        self.state = self._rng.uniform(-high, high)
        self.steps = 0
        return self._obs()

    def _obs(self):
        th, thdot = self.state
        return np.array(
            [math.cos(th), math.sin(th), thdot], dtype=np.float32
        )

    @staticmethod
    def _angle_normalize(x):
        return ((x + np.pi) % (2 * np.pi)) - np.pi

    def step(self, action):
        th, thdot = self.state
        u = float(np.clip(np.asarray(action).reshape(-1)[0],
                          -self.max_torque, self.max_torque))
        cost = (
            self._angle_normalize(th) ** 2 + 0.1 * thdot ** 2 + 0.001 * u ** 2
        )
        newthdot = thdot + (
            3 * self.g / (2 * self.l) * math.sin(th)
            + 3.0 / (self.m * self.l ** 2) * u
        ) * self.dt
        newthdot = float(np.clip(newthdot, -self.max_speed, self.max_speed))
        newth = th + newthdot * self.dt
        self.state = np.array([newth, newthdot])
        self.steps += 1
        done = self.steps >= self.max_episode_steps
        return self._obs(), -float(cost), bool(done), {}

    def render(self, *_, **__):
        return None

    def close(self):
        pass


def make(name: str, seed: Optional[int] = None):
    """Factory: 'CartPole-v1' / 'Pendulum-v1' / 'PixelCatch-v0'
    (version suffix ignored)."""
    base = name.split("-")[0].lower()
    if base == "cartpole":
        return CartPoleEnv(seed)
    if base == "pendulum":
        return PendulumEnv(seed)
    if base == "pixelcatch":
        from .pixel_catch import PixelCatchEnv

        return PixelCatchEnv(seed)
    raise ValueError(f"Unknown built-in environment {name!r}.")
