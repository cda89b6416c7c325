"""TRPO: trust-region policy optimization.

Parity target: reference ``machin/frame/algorithms/trpo.py``: flat-grad
machinery (:441-499), conjugate gradients (:304-339), Fisher-vector
products by double backward of the self-KL (:372-440), backtracking
line search against ``kl_max_delta`` (:340-371). Model contract: the
actor subclasses machin_amd.model.algorithms.trpo bases (``get_kl`` /
``compare_kl``).
"""
from typing import Dict

import torch as t
import torch.nn as nn

from .a2c import A2C
from .utils import safe_call


class TRPO(A2C):
    def __init__(
        self,
        actor: nn.Module,
        critic: nn.Module,
        optimizer,
        criterion,
        *_,
        kl_max_delta: float = 0.01,
        damping: float = 0.1,
        conjugate_iterations: int = 10,
        conjugate_res_threshold: float = 1e-10,
        line_search_backtracks: int = 10,
        **kwargs,
    ):
        super().__init__(actor, critic, optimizer, criterion, **kwargs)
        self.kl_max_delta = kl_max_delta
        self.damping = damping
        self.conjugate_iterations = conjugate_iterations
        self.conjugate_res_threshold = conjugate_res_threshold
        self.line_search_backtracks = line_search_backtracks

    # -- flat parameter/grad helpers -----------------------------------
    def _flat_params(self) -> t.Tensor:
        return t.cat([p.data.view(-1) for p in self.actor.parameters()])

    def _set_flat_params(self, flat: t.Tensor):
        offset = 0
        for p in self.actor.parameters():
            n = p.numel()
            p.data.copy_(flat[offset : offset + n].view_as(p))
            offset += n

    def _flat_grad(self, loss: t.Tensor, create_graph=False,
                   retain_graph=None) -> t.Tensor:
        grads = t.autograd.grad(
            loss,
            list(self.actor.parameters()),
            create_graph=create_graph,
            retain_graph=retain_graph,
            allow_unused=True,
        )
        flat = []
        for g, p in zip(grads, self.actor.parameters()):
            flat.append(
                g.contiguous().view(-1)
                if g is not None
                else t.zeros(p.numel(), device=p.device)
            )
        return t.cat(flat)

    # -- Fisher-vector product -----------------------------------------
    def _fvp(self, state: Dict, v: t.Tensor) -> t.Tensor:
        kl = safe_call(self.actor, state, method="get_kl")
        grad_kl = self._flat_grad(kl, create_graph=True)
        gvp = (grad_kl * v).sum()
        hvp = self._flat_grad(gvp, retain_graph=False)
        return hvp + self.damping * v

    def _conjugate_gradient(self, state: Dict, b: t.Tensor) -> t.Tensor:
        """Solve H x = b by CG."""
        x = t.zeros_like(b)
        r = b.clone()
        p = b.clone()
        rdotr = r.dot(r)
        for _ in range(self.conjugate_iterations):
            hp = self._fvp(state, p)
            alpha = rdotr / (p.dot(hp) + 1e-12)
            x += alpha * p
            r -= alpha * hp
            new_rdotr = r.dot(r)
            if new_rdotr < self.conjugate_res_threshold:
                break
            p = r + (new_rdotr / rdotr) * p
            rdotr = new_rdotr
        return x

    # -- update --------------------------------------------------------
    def update(self, update_value=True, update_policy=True,
               concatenate_samples=True, **__):
        (
            batch_size,
            (state, action, reward, next_state, terminal, target_value,
             advantage),
        ) = self.replay_buffer.sample_batch(
            -1,
            sample_method="all",
            concatenate=concatenate_samples,
            sample_attrs=[
                "state", "action", "reward", "next_state", "terminal",
                "value", "gae",
            ],
            additional_concat_custom_attrs=["value", "gae"],
        )
        if batch_size == 0:
            return 0.0, 0.0
        self.actor.train()
        self.critic.train()

        with t.no_grad():
            old_result = self._eval_act(state, action)
            old_log_prob = old_result[1].view(batch_size, 1)
            old_params_data = safe_call(
                self.actor, state, method="get_dist_params"
            )
            if isinstance(old_params_data, tuple):
                old_params_data = tuple(x.detach() for x in old_params_data)
            else:
                old_params_data = (old_params_data.detach(),)

        def surrogate_loss():
            result = self._eval_act(state, action)
            new_log_prob = result[1].view(batch_size, 1)
            adv = advantage.to(new_log_prob.device).view(batch_size, 1)
            if self.normalize_advantage:
                adv = (adv - adv.mean()) / (adv.std() + 1e-6)
            ratio = (new_log_prob - old_log_prob).exp()
            return -(ratio * adv).mean()

        loss = surrogate_loss()
        g = self._flat_grad(loss, retain_graph=False)
        if update_policy and g.abs().max() > 0:
            step_dir = self._conjugate_gradient(state, -g)
            shs = 0.5 * step_dir.dot(self._fvp(state, step_dir))
            if shs.item() > 0:
                lm = (shs / self.kl_max_delta).sqrt()
                full_step = step_dir / lm
                expected_improve = -g.dot(full_step)

                prev_params = self._flat_params()
                success = False
                frac = 1.0
                for _ in range(self.line_search_backtracks):
                    self._set_flat_params(prev_params + frac * full_step)
                    if len(old_params_data) == 1:
                        old_kwargs = {"old_logits": old_params_data[0]}
                    else:
                        old_kwargs = {
                            "old_mean": old_params_data[0],
                            "old_log_std": old_params_data[1],
                        }
                    with t.no_grad():
                        new_loss = surrogate_loss()
                        kl = safe_call(
                            self.actor, old_kwargs, state, method="compare_kl"
                        )
                    improve = loss - new_loss
                    if (
                        kl.item() <= self.kl_max_delta * 1.5
                        and improve.item() > 0
                        and improve.item()
                        > 0.1 * frac * expected_improve.item()
                    ):
                        success = True
                        break
                    frac *= 0.5
                if not success:
                    self._set_flat_params(prev_params)

        # critic updates (same as A2C)
        sum_value_loss = 0.0
        for _ in range(self.critic_update_times):
            value = self._criticize(state).view(batch_size, 1)
            tv = target_value.to(value.device).view(batch_size, 1)
            value_loss = (
                self.criterion(value, tv.to(value.dtype)) * self.value_weight
            )
            if update_value:
                self.critic_optim.zero_grad(set_to_none=True)
                self._backward(value_loss)
                nn.utils.clip_grad_norm_(
                    self.critic.parameters(), self.grad_max
                )
                self.critic_optim.step()
            sum_value_loss += float(value_loss.detach().item())

        self.replay_buffer.clear()
        return (
            -float(loss.detach().item()),
            sum_value_loss / max(self.critic_update_times, 1),
        )

    @classmethod
    def generate_config(cls, config):
        config = A2C.generate_config(config)
        config["frame"] = "TRPO"
        fc = config["frame_config"]
        fc["frame"] = "TRPO"
        fc.setdefault("kl_max_delta", 0.01)
        fc.setdefault("damping", 0.1)
        fc.setdefault("conjugate_iterations", 10)
        fc.setdefault("line_search_backtracks", 10)
        return config
