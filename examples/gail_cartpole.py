"""GAIL on CartPole: train an expert with PPO, collect
demonstrations, imitate them with GAIL (reference analog:
examples/framework_examples/gail.py)."""
import os
import sys

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import torch as t
import torch.nn as nn

from machin_amd.env.envs import CartPoleEnv
from machin_amd.frame.algorithms import GAIL, PPO


class Actor(nn.Module):
    def __init__(self, state_dim=4, action_num=2):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, 64)
        self.fc2 = nn.Linear(64, action_num)

    def forward(self, state, action=None):
        logits = self.fc2(t.relu(self.fc1(state)))
        dist = t.distributions.Categorical(logits=logits)
        if action is None:
            action = dist.sample().view(-1, 1)
        return (action, dist.log_prob(action.view(-1)).view(-1, 1),
                dist.entropy().view(-1, 1))


class Critic(nn.Module):
    def __init__(self, state_dim=4):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, 64)
        self.fc2 = nn.Linear(64, 1)

    def forward(self, state):
        return self.fc2(t.relu(self.fc1(state)))


class Discriminator(nn.Module):
    def __init__(self, state_dim=4, action_dim=1):
        super().__init__()
        self.fc1 = nn.Linear(state_dim + action_dim, 64)
        self.fc2 = nn.Linear(64, 1)

    def forward(self, state, action):
        x = t.cat([state, action.float()], dim=1)
        return t.sigmoid(self.fc2(t.relu(self.fc1(x))))


def make_ppo():
    return PPO(
        Actor(), Critic(), t.optim.Adam, nn.MSELoss(),
        entropy_weight=0.01, gae_lambda=0.97,
        actor_learning_rate=2e-3, critic_learning_rate=2e-3,
        actor_update_times=6, critic_update_times=10,
    )


def run_episode(env, act_fn):
    obs = t.tensor(env.reset()).view(1, 4)
    total, transitions, done = 0.0, [], False
    while not done:
        with t.no_grad():
            action = act_fn(obs)
        o, r, done, _ = env.step(int(action.item()))
        o = t.tensor(o).view(1, 4)
        total += r
        transitions.append(
            {"state": {"state": obs}, "action": {"action": action},
             "next_state": {"state": o}, "reward": r,
             "terminal": done and env.steps < env.max_episode_steps}
        )
        obs = o
    return total, transitions


def main():
    env = CartPoleEnv(seed=0)
    # 1) train an expert
    expert = make_ppo()
    smoothed = 0.0
    for episode in range(1000):
        total, transitions = run_episode(
            env, lambda s: expert.act({"state": s})[0]
        )
        expert.store_episode(transitions)
        expert.update()
        smoothed = smoothed * 0.9 + total * 0.1
        if smoothed > 195:
            print(f"expert ready at episode {episode}")
            break
    # 2) collect demonstrations (state-action only)
    demos = []
    for _ in range(10):
        _, transitions = run_episode(
            env, lambda s: expert.act({"state": s})[0]
        )
        demos.append(
            [{"state": tr["state"], "action": tr["action"]}
             for tr in transitions]
        )
    # 3) imitate with GAIL (env reward replaced by the discriminator)
    gail = GAIL(Discriminator(), make_ppo(), t.optim.Adam,
                discriminator_learning_rate=1e-3)
    for demo in demos:
        gail.store_expert_episode(demo)
    smoothed = 0.0
    for episode in range(1000):
        total, transitions = run_episode(
            env, lambda s: gail.act({"state": s})[0]
        )
        gail.store_episode(transitions)
        gail.update()
        smoothed = smoothed * 0.9 + total * 0.1
        if episode % 20 == 0:
            print(f"gail episode {episode}: smoothed env reward "
                  f"{smoothed:.1f}")
        if smoothed > 195:
            print(f"gail solved at episode {episode}")
            return
    print("gail did not reach 195 within the budget")


if __name__ == "__main__":
    main()
