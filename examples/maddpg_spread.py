"""MADDPG on cooperative navigation (reference analog:
examples/framework_examples/maddpg.py on MPE simple_spread)."""
import os
import sys

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import argparse

import numpy as np
import torch as t
import torch.nn as nn

from machin_amd.env.envs import SimpleSpreadEnv
from machin_amd.frame.algorithms import MADDPG

N = 3
OBS = 4 + 2 * N + 2 * (N - 1)


class Actor(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(OBS, 64)
        self.fc2 = nn.Linear(64, 64)
        self.fc3 = nn.Linear(64, 2)

    def forward(self, state):
        a = t.relu(self.fc1(state))
        return t.tanh(self.fc3(t.relu(self.fc2(a))))


class Critic(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(N * (OBS + 2), 64)
        self.fc2 = nn.Linear(64, 64)
        self.fc3 = nn.Linear(64, 1)

    def forward(self, state, action):
        x = t.cat([state, action], dim=1)
        return self.fc3(t.relu(self.fc2(t.relu(self.fc1(x)))))


def main(episodes=400):
    maddpg = MADDPG(
        [Actor() for _ in range(N)], [Actor() for _ in range(N)],
        [Critic() for _ in range(N)], [Critic() for _ in range(N)],
        t.optim.Adam, nn.MSELoss(),
        batch_size=256, update_rate=0.01,
        actor_learning_rate=1e-3, critic_learning_rate=2e-3,
    )
    env = SimpleSpreadEnv(n_agents=N, seed=0)
    smoothed = None
    for episode in range(episodes):
        obs = env.reset()
        noise = max(0.4 * (1 - episode / 300), 0.05)
        total, done = 0.0, False
        agent_eps = [[] for _ in range(N)]
        while not done:
            states = [{"state": t.tensor(o).view(1, -1)} for o in obs]
            with t.no_grad():
                actions = maddpg.act_with_noise(
                    states, noise_param=(0.0, noise), mode="normal"
                )
            actions = [a.clamp(-1, 1) for a in actions]
            obs2, rewards, done, _ = env.step(
                [a.view(-1).numpy() for a in actions]
            )
            total += rewards[0]
            for i in range(N):
                agent_eps[i].append(
                    {"state": states[i],
                     "action": {"action": actions[i].view(1, 2)},
                     "next_state": {"state": t.tensor(obs2[i]).view(1, -1)},
                     "reward": rewards[i] / 10.0, "terminal": False}
                )
            obs = obs2
        maddpg.store_episodes(agent_eps)
        if maddpg.replay_buffers[0].size() > 1000:
            for _ in range(5):
                maddpg.update()
        smoothed = total if smoothed is None else smoothed * 0.95 + total * 0.05
        if episode % 20 == 0:
            print(f"episode {episode}: smoothed shared reward {smoothed:.2f}")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--episodes", type=int, default=400)
    main(p.parse_args().episodes)
