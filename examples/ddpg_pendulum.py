"""DDPG on Pendulum (reference analog:
examples/framework_examples/ddpg.py)."""
import os
import sys

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import argparse

import torch as t
import torch.nn as nn

from machin_amd.env.envs import PendulumEnv
from machin_amd.frame.algorithms import DDPG


class Actor(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(3, 64)
        self.fc2 = nn.Linear(64, 64)
        self.fc3 = nn.Linear(64, 1)

    def forward(self, state):
        a = t.relu(self.fc2(t.relu(self.fc1(state))))
        return t.tanh(self.fc3(a)) * 2.0


class Critic(nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(4, 64)
        self.fc2 = nn.Linear(64, 64)
        self.fc3 = nn.Linear(64, 1)

    def forward(self, state, action):
        x = t.cat([state, action], dim=1)
        return self.fc3(t.relu(self.fc2(t.relu(self.fc1(x)))))


def main(device="cpu", max_episodes=300):
    ddpg = DDPG(
        Actor().to(device), Actor().to(device),
        Critic().to(device), Critic().to(device),
        t.optim.Adam, nn.MSELoss(),
        batch_size=100, update_rate=0.005, replay_device=device,
        actor_learning_rate=5e-4, critic_learning_rate=1e-3,
    )
    env = PendulumEnv(seed=0)
    smoothed = -1600.0
    for episode in range(max_episodes):
        obs = t.tensor(env.reset(), device=device).view(1, 3)
        total, transitions, done = 0.0, [], False
        while not done:
            with t.no_grad():
                action = ddpg.act_with_noise(
                    {"state": obs}, noise_param=(0.0, 0.3), mode="normal"
                ).clamp(-2, 2)
            o, r, done, _ = env.step(action.view(-1).cpu().numpy())
            o = t.tensor(o, device=device).view(1, 3)
            total += r
            transitions.append(
                {"state": {"state": obs},
                 "action": {"action": action.view(1, 1)},
                 "next_state": {"state": o}, "reward": r / 10.0,
                 "terminal": False}
            )
            obs = o
        ddpg.store_episode(transitions)
        if ddpg.replay_buffer.size() > 500:
            for _ in range(100):
                ddpg.update()
        smoothed = smoothed * 0.9 + total * 0.1
        if episode % 10 == 0:
            print(f"episode {episode}: smoothed reward {smoothed:.1f}")
        if smoothed > -300:
            print(f"solved at episode {episode}")
            return
    print("did not reach -300 within the budget")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cpu")
    main(p.parse_args().device)
