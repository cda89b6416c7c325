"""TRPO actor base classes.

Parity target: reference ``machin/model/algorithms/trpo.py`` (:8-149):
categorical and diagonal-Gaussian actor bases exposing the TRPO model
contract — ``forward(state[, action]) -> (action, log_prob, entropy)``,
``get_kl`` (self-KL whose Hessian is the Fisher matrix) and
``compare_kl`` (KL between a frozen old distribution and the current
one).

Users subclass and implement the distribution-parameter head:
``policy_logits(...)`` for the discrete base, ``policy_mean_std(...)``
for the continuous base. State keyword names must match the
``forward`` signature of the subclass head.
"""
from typing import Dict

import torch as t

from ..nets.base import NeuralNetworkModule


class TRPOActorDiscrete(NeuralNetworkModule):
    """Categorical-policy base."""

    def policy_logits(self, *args, **kwargs) -> t.Tensor:
        """Subclass: return action logits [batch, action_num]."""
        raise NotImplementedError

    def forward(self, *args, action: t.Tensor = None, **kwargs):
        logits = self.policy_logits(*args, **kwargs)
        dist = t.distributions.Categorical(logits=logits)
        if action is None:
            action = dist.sample().view(-1, 1)
        log_prob = dist.log_prob(action.view(-1)).view(-1, 1)
        entropy = dist.entropy().view(-1, 1)
        return action, log_prob, entropy

    def get_kl(self, *args, **kwargs) -> t.Tensor:
        """Mean KL(detach(pi) || pi): zero value, Fisher Hessian."""
        logits = self.policy_logits(*args, **kwargs)
        log_p = t.log_softmax(logits, dim=-1)
        log_p0 = log_p.detach()
        p0 = log_p0.exp()
        return (p0 * (log_p0 - log_p)).sum(dim=-1).mean()

    def compare_kl(self, old_logits: t.Tensor, *args, **kwargs) -> t.Tensor:
        """Mean KL(old || current) on the same states."""
        logits = self.policy_logits(*args, **kwargs)
        log_p = t.log_softmax(logits, dim=-1)
        log_p0 = t.log_softmax(old_logits, dim=-1)
        return (log_p0.exp() * (log_p0 - log_p)).sum(dim=-1).mean()

    def get_dist_params(self, *args, **kwargs) -> t.Tensor:
        return self.policy_logits(*args, **kwargs)


class TRPOActorContinuous(NeuralNetworkModule):
    """Diagonal-Gaussian policy base."""

    def policy_mean_std(self, *args, **kwargs):
        """Subclass: return (mean [B,D], log_std [B,D] or [D])."""
        raise NotImplementedError

    def _dist(self, *args, **kwargs):
        mean, log_std = self.policy_mean_std(*args, **kwargs)
        if log_std.dim() < mean.dim():
            log_std = log_std.expand_as(mean)
        return mean, log_std

    def forward(self, *args, action: t.Tensor = None, **kwargs):
        mean, log_std = self._dist(*args, **kwargs)
        dist = t.distributions.Normal(mean, log_std.exp())
        if action is None:
            action = dist.sample()
        log_prob = dist.log_prob(action).sum(dim=-1, keepdim=True)
        entropy = dist.entropy().sum(dim=-1, keepdim=True)
        return action, log_prob, entropy

    def get_kl(self, *args, **kwargs) -> t.Tensor:
        mean, log_std = self._dist(*args, **kwargs)
        m0, ls0 = mean.detach(), log_std.detach()
        var, var0 = (2 * log_std).exp(), (2 * ls0).exp()
        kl = (
            log_std - ls0
            + (var0 + (m0 - mean) ** 2) / (2.0 * var)
            - 0.5
        )
        return kl.sum(dim=-1).mean()

    def compare_kl(self, old_mean: t.Tensor, old_log_std: t.Tensor,
                   *args, **kwargs) -> t.Tensor:
        mean, log_std = self._dist(*args, **kwargs)
        var = (2 * log_std).exp()
        var0 = (2 * old_log_std).exp()
        kl = (
            log_std - old_log_std
            + (var0 + (old_mean - mean) ** 2) / (2.0 * var)
            - 0.5
        )
        return kl.sum(dim=-1).mean()

    def get_dist_params(self, *args, **kwargs):
        return self._dist(*args, **kwargs)
