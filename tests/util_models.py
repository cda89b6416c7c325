"""Shared tiny models for algorithm tests."""
import torch as t
import torch.nn as nn

from machin_amd.model.algorithms.trpo import (
    TRPOActorContinuous,
    TRPOActorDiscrete,
)


class QNet(nn.Module):
    def __init__(self, state_dim=4, action_num=2):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, 16)
        self.fc2 = nn.Linear(16, 16)
        self.fc3 = nn.Linear(16, action_num)

    def forward(self, state):
        a = t.relu(self.fc1(state))
        a = t.relu(self.fc2(a))
        return self.fc3(a)


class DistQNet(nn.Module):
    """C51 network: [B, action_num, atoms] distribution."""

    def __init__(self, state_dim=4, action_num=2, atom_num=11):
        super().__init__()
        self.action_num = action_num
        self.atom_num = atom_num
        self.fc1 = nn.Linear(state_dim, 32)
        self.fc2 = nn.Linear(32, action_num * atom_num)

    def forward(self, state):
        a = t.relu(self.fc1(state))
        a = self.fc2(a).view(-1, self.action_num, self.atom_num)
        return t.softmax(a, dim=-1)


class DetActor(nn.Module):
    """Deterministic tanh actor (DDPG family)."""

    def __init__(self, state_dim=3, action_dim=1, action_range=2.0):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, 16)
        self.fc2 = nn.Linear(16, 16)
        self.fc3 = nn.Linear(16, action_dim)
        self.action_range = action_range

    def forward(self, state):
        a = t.relu(self.fc1(state))
        a = t.relu(self.fc2(a))
        return t.tanh(self.fc3(a)) * self.action_range


class Critic(nn.Module):
    """Q(s, a) critic."""

    def __init__(self, state_dim=3, action_dim=1):
        super().__init__()
        self.fc1 = nn.Linear(state_dim + action_dim, 16)
        self.fc2 = nn.Linear(16, 16)
        self.fc3 = nn.Linear(16, 1)

    def forward(self, state, action):
        x = t.cat([state, action], dim=1)
        x = t.relu(self.fc1(x))
        x = t.relu(self.fc2(x))
        return self.fc3(x)


class VCritic(nn.Module):
    """V(s) critic (A2C family)."""

    def __init__(self, state_dim=4):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, 16)
        self.fc2 = nn.Linear(16, 16)
        self.fc3 = nn.Linear(16, 1)

    def forward(self, state):
        x = t.relu(self.fc1(state))
        x = t.relu(self.fc2(x))
        return self.fc3(x)


class StochDiscreteActor(nn.Module):
    """Categorical actor for A2C/PPO: (action, log_prob, entropy)."""

    def __init__(self, state_dim=4, action_num=2):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, 16)
        self.fc2 = nn.Linear(16, 16)
        self.fc3 = nn.Linear(16, action_num)

    def forward(self, state, action=None):
        a = t.relu(self.fc1(state))
        a = t.relu(self.fc2(a))
        logits = self.fc3(a)
        dist = t.distributions.Categorical(logits=logits)
        if action is None:
            action = dist.sample().view(-1, 1)
        log_prob = dist.log_prob(action.view(-1)).view(-1, 1)
        entropy = dist.entropy().view(-1, 1)
        return action, log_prob, entropy


class GaussianActor(nn.Module):
    """Tanh-squashed gaussian actor for SAC:
    (action, log_prob)."""

    def __init__(self, state_dim=3, action_dim=1, action_range=2.0):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, 16)
        self.fc2 = nn.Linear(16, 16)
        self.mu = nn.Linear(16, action_dim)
        self.log_std = nn.Linear(16, action_dim)
        self.action_range = action_range

    def forward(self, state):
        x = t.relu(self.fc1(state))
        x = t.relu(self.fc2(x))
        mu = self.mu(x)
        log_std = self.log_std(x).clamp(-20, 2)
        dist = t.distributions.Normal(mu, log_std.exp())
        u = dist.rsample()
        a = t.tanh(u)
        log_prob = (
            dist.log_prob(u) - t.log(1 - a.pow(2) + 1e-6)
        ).sum(dim=1, keepdim=True)
        return a * self.action_range, log_prob


class TRPODiscreteActor(TRPOActorDiscrete):
    def __init__(self, state_dim=4, action_num=2):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, 16)
        self.fc2 = nn.Linear(16, action_num)

    def policy_logits(self, state):
        return self.fc2(t.relu(self.fc1(state)))


class TRPOGaussianActor(TRPOActorContinuous):
    def __init__(self, state_dim=3, action_dim=1):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, 16)
        self.mu = nn.Linear(16, action_dim)
        self.log_std_param = nn.Parameter(t.zeros(action_dim))

    def policy_mean_std(self, state):
        h = t.relu(self.fc1(state))
        return self.mu(h), self.log_std_param.expand(state.shape[0], -1)


class Discriminator(nn.Module):
    def __init__(self, state_dim=4, action_dim=1):
        super().__init__()
        self.fc1 = nn.Linear(state_dim + action_dim, 16)
        self.fc2 = nn.Linear(16, 1)

    def forward(self, state, action):
        x = t.cat([state, action.float()], dim=1)
        return t.sigmoid(self.fc2(t.relu(self.fc1(x))))
