"""Cooperative navigation ("simple spread") environment.

Stand-in for the MPE ``simple_spread`` scenario the reference's MADDPG
CI trains on (machin test_lib/multiagent-particle-envs;
test/frame/algorithms/test_maddpg.py:80-97): N point-mass agents must
cover N landmarks; the shared reward is the negative sum over
landmarks of the distance to the closest agent, minus a collision
penalty. Continuous 2-D force actions in [-1, 1].

Per-agent observation: [own vel(2), own pos(2),
landmark deltas(2N), other-agent deltas(2(N-1))].
"""
from typing import List, Optional

import numpy as np


class SimpleSpreadEnv:
    max_episode_steps = 25

    def __init__(self, n_agents: int = 3, seed: Optional[int] = None):
        self.n = n_agents
        self.dt = 0.1
        self.damping = 0.75
        self.accel = 5.0
        self.max_speed = 1.3
        self.agent_size = 0.15
        self._rng = np.random.RandomState(seed)
        self.obs_dim = 4 + 2 * self.n + 2 * (self.n - 1)
        self.action_dim = 2
        self.pos = None
        self.vel = None
        self.landmarks = None
        self.steps = 0

    def seed(self, seed=None):
        self._rng = np.random.RandomState(seed)
        return [seed]

    def reset(self) -> List[np.ndarray]:
        self.pos = self._rng.uniform(-1, 1, size=(self.n, 2))
        self.vel = np.zeros((self.n, 2))
        self.landmarks = self._rng.uniform(-1, 1, size=(self.n, 2))
        self.steps = 0
        return self._observe()

    def _observe(self) -> List[np.ndarray]:
        obs = []
        for i in range(self.n):
            parts = [self.vel[i], self.pos[i]]
            for lm in self.landmarks:
                parts.append(lm - self.pos[i])
            for j in range(self.n):
                if j != i:
                    parts.append(self.pos[j] - self.pos[i])
            obs.append(np.concatenate(parts).astype(np.float32))
        return obs

    def step(self, actions: List[np.ndarray]):
        """actions: one [2] array in [-1, 1] per agent. Returns
        (observations, rewards, done, info); the reward is shared."""
        acts = np.clip(np.asarray(actions, dtype=np.float64), -1, 1)
        acts = acts.reshape(self.n, 2)
        self.vel = self.vel * self.damping + acts * self.accel * self.dt
        speed = np.linalg.norm(self.vel, axis=1, keepdims=True)
        too_fast = speed > self.max_speed
        self.vel = np.where(
            too_fast, self.vel / np.maximum(speed, 1e-8) * self.max_speed,
            self.vel,
        )
        self.pos = self.pos + self.vel * self.dt

        # shared reward: coverage + collision penalty
        reward = 0.0
        for lm in self.landmarks:
            dists = np.linalg.norm(self.pos - lm, axis=1)
            reward -= float(dists.min())
        for i in range(self.n):
            for j in range(i + 1, self.n):
                if (
                    np.linalg.norm(self.pos[i] - self.pos[j])
                    < 2 * self.agent_size
                ):
                    reward -= 1.0

        self.steps += 1
        done = self.steps >= self.max_episode_steps
        rewards = [reward] * self.n
        return self._observe(), rewards, done, {}

    def render(self, *_, **__):
        return None

    def close(self):
        pass
