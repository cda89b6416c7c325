"""Action-space noise adders.

Parity target: reference ``machin/frame/noise/action_space_noise.py``
(:12-171): four modes — uniform, normal, clipped-normal, OU — each
taking per-call (or per-dimension tuples of) noise parameters.
"""
from typing import Any, Tuple, Union

import torch as t

from .generator import (
    ClippedNormalNoiseGen,
    NormalNoiseGen,
    OrnsteinUhlenbeckNoiseGen,
    UniformNoiseGen,
)


def _apply_per_dim(action: t.Tensor, gen_cls, noise_param) -> t.Tensor:
    """noise_param is a tuple of per-dim tuples: build noise per dim."""
    noise = t.zeros_like(action)
    for dim, params in enumerate(noise_param):
        gen = gen_cls((action.shape[0], 1), *params)
        noise[:, dim : dim + 1] = gen(action.device)
    return action + noise


def add_uniform_noise_to_action(
    action: t.Tensor, noise_param: Union[Tuple, Any] = (0.0, 1.0),
    ratio: float = 1.0,
):
    """Add uniform noise. ``noise_param=(min,max)`` or a tuple of
    per-dimension ``(min,max)`` tuples."""
    if isinstance(noise_param[0], (tuple, list)):
        noise = t.zeros_like(action)
        for dim, (lo, hi) in enumerate(noise_param):
            noise[:, dim : dim + 1] = (
                t.rand((action.shape[0], 1), device=action.device) * (hi - lo)
                + lo
            )
        return action + noise * ratio
    lo, hi = noise_param
    return (
        action
        + (t.rand_like(action) * (hi - lo) + lo) * ratio
    )


def add_normal_noise_to_action(
    action: t.Tensor, noise_param=(0.0, 1.0), ratio: float = 1.0
):
    """Add gaussian noise. ``noise_param=(mu,sigma)`` or per-dim tuples."""
    if isinstance(noise_param[0], (tuple, list)):
        noise = t.zeros_like(action)
        for dim, (mu, sigma) in enumerate(noise_param):
            noise[:, dim : dim + 1] = (
                t.randn((action.shape[0], 1), device=action.device) * sigma
                + mu
            )
        return action + noise * ratio
    mu, sigma = noise_param
    return action + (t.randn_like(action) * sigma + mu) * ratio


def add_clipped_normal_noise_to_action(
    action: t.Tensor, noise_param=(0.0, 1.0, -1.0, 1.0), ratio: float = 1.0
):
    """Add clipped gaussian noise. ``noise_param=(mu,sigma,min,max)``."""
    if isinstance(noise_param[0], (tuple, list)):
        noise = t.zeros_like(action)
        for dim, (mu, sigma, nmin, nmax) in enumerate(noise_param):
            noise[:, dim : dim + 1] = (
                t.randn((action.shape[0], 1), device=action.device) * sigma
                + mu
            ).clamp(nmin, nmax)
        return action + noise * ratio
    mu, sigma, nmin, nmax = noise_param
    return (
        action
        + (t.randn_like(action) * sigma + mu).clamp(nmin, nmax) * ratio
    )


def add_ou_noise_to_action(
    action: t.Tensor, noise_param: dict = None, ratio: float = 1.0,
    reset: bool = False,
):
    """Add Ornstein-Uhlenbeck noise. ``noise_param`` is a dict of
    OrnsteinUhlenbeckNoiseGen kwargs; the generator persists across
    calls (temporal correlation) until ``reset=True``."""
    global _ou_gen
    noise_param = noise_param or {}
    if "_ou_gen" not in globals() or _ou_gen is None or reset:
        _ou_gen = OrnsteinUhlenbeckNoiseGen(action.shape, **noise_param)
    return action + _ou_gen(action.device) * ratio


_ou_gen = None
