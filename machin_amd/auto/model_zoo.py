"""Default models for auto-generated configs (resolved by import
path ``machin_amd.auto.model_zoo.<Name>``)."""
import torch as t
import torch.nn as nn


class QNet(nn.Module):
    def __init__(self, state_dim=4, action_num=2, hidden=64):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, hidden)
        self.fc2 = nn.Linear(hidden, hidden)
        self.fc3 = nn.Linear(hidden, action_num)

    def forward(self, state):
        a = t.relu(self.fc1(state))
        a = t.relu(self.fc2(a))
        return self.fc3(a)


class DistQNet(nn.Module):
    def __init__(self, state_dim=4, action_num=2, atom_num=51, hidden=64):
        super().__init__()
        self.action_num = action_num
        self.atom_num = atom_num
        self.fc1 = nn.Linear(state_dim, hidden)
        self.fc2 = nn.Linear(hidden, action_num * atom_num)

    def forward(self, state):
        a = t.relu(self.fc1(state))
        a = self.fc2(a).view(-1, self.action_num, self.atom_num)
        return t.softmax(a, dim=-1)


class StochasticActor(nn.Module):
    """Categorical actor: (action, log_prob, entropy)."""

    def __init__(self, state_dim=4, action_num=2, hidden=64):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, hidden)
        self.fc2 = nn.Linear(hidden, action_num)

    def forward(self, state, action=None):
        logits = self.fc2(t.relu(self.fc1(state)))
        if action is None:
            dist = t.distributions.Categorical(logits=logits)
            action = dist.sample().view(-1, 1)
            log_prob = dist.log_prob(action.view(-1)).view(-1, 1)
            entropy = dist.entropy().view(-1, 1)
            return action, log_prob, entropy
        # evaluation of GIVEN actions (the A2C/PPO/IMPALA update
        # path): fused log_softmax+gather+entropy kernel on ROCm,
        # identical torch math on CPU
        from .. import ops

        log_prob, entropy = ops.categorical_policy_head(
            logits, action.view(-1)
        )
        return action, log_prob.view(-1, 1), entropy.view(-1, 1)


class DeterministicActor(nn.Module):
    def __init__(self, state_dim=3, action_dim=1, action_range=2.0,
                 hidden=64):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, hidden)
        self.fc2 = nn.Linear(hidden, action_dim)
        self.action_range = action_range

    def forward(self, state):
        return t.tanh(self.fc2(t.relu(self.fc1(state)))) * self.action_range


class GaussianActor(nn.Module):
    """Tanh-squashed Gaussian: (action, log_prob). Sampling + log-prob
    run through machin_amd.model.nets.GaussianPolicyHead — the fused
    gfx950 kernel on ROCm, identical torch math on CPU."""

    def __init__(self, state_dim=3, action_dim=1, action_range=2.0,
                 hidden=64):
        super().__init__()
        from ..model.nets.gaussian import GaussianPolicyHead

        self.fc1 = nn.Linear(state_dim, hidden)
        self.mu = nn.Linear(hidden, action_dim)
        self.log_std = nn.Linear(hidden, action_dim)
        self.head = GaussianPolicyHead(
            tanh_squash=True, action_range=action_range
        )

    def forward(self, state):
        h = t.relu(self.fc1(state))
        return self.head(self.mu(h), self.log_std(h))


class QCritic(nn.Module):
    def __init__(self, state_dim=3, action_dim=1, hidden=64):
        super().__init__()
        self.fc1 = nn.Linear(state_dim + action_dim, hidden)
        self.fc2 = nn.Linear(hidden, 1)

    def forward(self, state, action):
        return self.fc2(t.relu(self.fc1(t.cat([state, action], dim=1))))


class VCritic(nn.Module):
    def __init__(self, state_dim=4, hidden=64):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, hidden)
        self.fc2 = nn.Linear(hidden, 1)

    def forward(self, state):
        return self.fc2(t.relu(self.fc1(state)))


class TRPOStochasticActor(nn.Module):
    """Categorical actor with the TRPO model contract."""

    def __init__(self, state_dim=4, action_num=2, hidden=64):
        super().__init__()
        self.fc1 = nn.Linear(state_dim, hidden)
        self.fc2 = nn.Linear(hidden, action_num)

    def policy_logits(self, state):
        return self.fc2(t.relu(self.fc1(state)))

    def forward(self, state, action=None):
        logits = self.policy_logits(state)
        dist = t.distributions.Categorical(logits=logits)
        if action is None:
            action = dist.sample().view(-1, 1)
        log_prob = dist.log_prob(action.view(-1)).view(-1, 1)
        entropy = dist.entropy().view(-1, 1)
        return action, log_prob, entropy

    def get_kl(self, state):
        logits = self.policy_logits(state)
        log_p = t.log_softmax(logits, dim=-1)
        log_p0 = log_p.detach()
        return (log_p0.exp() * (log_p0 - log_p)).sum(dim=-1).mean()

    def compare_kl(self, old_logits, state):
        logits = self.policy_logits(state)
        log_p = t.log_softmax(logits, dim=-1)
        log_p0 = t.log_softmax(old_logits, dim=-1)
        return (log_p0.exp() * (log_p0 - log_p)).sum(dim=-1).mean()

    def get_dist_params(self, state):
        return self.policy_logits(state)


class ArgmaxActor(nn.Module):
    """ARS-style deterministic discrete policy."""

    def __init__(self, state_dim=4, action_num=2):
        super().__init__()
        self.fc = nn.Linear(state_dim, action_num, bias=False)

    def forward(self, state):
        return t.argmax(self.fc(state), dim=1)
