"""TRPO on CartPole (reference analog:
examples/framework_examples/trpo.py)."""
import os
import sys

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import argparse

import torch as t
import torch.nn as nn

from machin_amd.env.envs import CartPoleEnv
from machin_amd.frame.algorithms import TRPO
from machin_amd.model.algorithms.trpo import TRPOActorDiscrete


class Actor(TRPOActorDiscrete):
    """Categorical policy with the TRPO model contract
    (get_kl/compare_kl provided by the base)."""

    def __init__(self, state_dim=4, action_num=2):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, 64)
        self.fc2 = nn.Linear(64, action_num)

    def policy_logits(self, state):
        return self.fc2(t.relu(self.fc1(state)))


class Critic(nn.Module):
    def __init__(self, state_dim=4):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, 64)
        self.fc2 = nn.Linear(64, 1)

    def forward(self, state):
        return self.fc2(t.relu(self.fc1(state)))


def main(device="cpu", max_episodes=1200):
    trpo = TRPO(
        Actor().to(device), Critic().to(device), t.optim.Adam,
        nn.MSELoss(), gae_lambda=0.97, critic_learning_rate=2e-3,
        kl_max_delta=0.01, replay_device=device,
    )
    env = CartPoleEnv(seed=0)
    smoothed = 0.0
    for episode in range(max_episodes):
        obs = t.tensor(env.reset(), device=device).view(1, 4)
        total, transitions, done = 0.0, [], False
        while not done:
            with t.no_grad():
                action = trpo.act({"state": obs})[0]
            o, r, done, _ = env.step(int(action.item()))
            o = t.tensor(o, device=device).view(1, 4)
            total += r
            transitions.append(
                {"state": {"state": obs}, "action": {"action": action},
                 "next_state": {"state": o}, "reward": r,
                 "terminal": done and env.steps < env.max_episode_steps}
            )
            obs = o
        trpo.store_episode(transitions)
        trpo.update()
        smoothed = smoothed * 0.9 + total * 0.1
        if episode % 20 == 0:
            print(f"episode {episode}: smoothed reward {smoothed:.1f}")
        if smoothed > 195:
            print(f"solved at episode {episode}")
            return
    print("did not reach 195 within the budget")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cpu")
    main(p.parse_args().device)
