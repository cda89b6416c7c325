"""Parameter-space noise with adaptive sigma.

Parity target: reference ``machin/frame/noise/param_space_noise.py``
(:10-132): ``AdaptiveParamNoise`` sigma adaptation and
``perturb_model`` — forward hooks that swap perturbed parameters in
for acting and restore the originals afterwards, plus a distance
callback to adapt sigma toward a target action-space distance.
"""
from typing import Callable

import torch as t
import torch.nn as nn


class AdaptiveParamNoise:
    """Sigma controller: grow sigma when perturbed-vs-clean action
    distance is below target, shrink when above."""

    def __init__(self, initial_stddev: float = 0.1,
                 desired_action_stddev: float = 0.1,
                 adoption_coefficient: float = 1.01):
        self.initial_stddev = initial_stddev
        self.desired_action_stddev = desired_action_stddev
        self.adoption_coefficient = adoption_coefficient
        self.current_stddev = initial_stddev

    def adapt(self, distance: float):
        if distance > self.desired_action_stddev:
            self.current_stddev /= self.adoption_coefficient
        else:
            self.current_stddev *= self.adoption_coefficient

    def get_dev(self) -> float:
        return self.current_stddev

    def __repr__(self):
        return (
            f"AdaptiveParamNoise(initial_stddev={self.initial_stddev}, "
            f"desired_action_stddev={self.desired_action_stddev}, "
            f"adoption_coefficient={self.adoption_coefficient})"
        )


def perturb_model(
    model: nn.Module,
    perturb_switch,
    reset_switch,
    distance_func: Callable = None,
    desired_action_stddev: float = 0.5,
    noise_generator=None,
    debug_backward: bool = False,
):
    """Install parameter-noise hooks on ``model``.

    While ``perturb_switch`` is on, forward passes run with perturbed
    parameters (originals restored right after); when ``reset_switch``
    is on, fresh noise is drawn at the next forward. Returns
    ``(cancel_hook_fn, param_noise_spec)``.

    ``distance_func(clean_output, perturbed_output) -> float`` feeds
    sigma adaptation; defaults to mean L2 distance.
    """
    spec = AdaptiveParamNoise(desired_action_stddev=desired_action_stddev)
    state = {"noise": None, "orig": None, "perturbed_out": None,
             "clean_out": None}

    if distance_func is None:
        def distance_func(clean, perturbed):
            return float((clean - perturbed).pow(2).mean().sqrt())

    def make_noise():
        dev = spec.get_dev()
        noise = {}
        for name, p in model.named_parameters():
            if noise_generator is not None:
                n = noise_generator(p.shape, dev)(p.device)
            else:
                n = t.randn_like(p) * dev
            noise[name] = n
        return noise

    def pre_hook(module, inputs):
        if perturb_switch.get():
            if state["noise"] is None or reset_switch.get():
                state["noise"] = make_noise()
                if hasattr(reset_switch, "off"):
                    reset_switch.off()
            state["orig"] = {
                n: p.data.clone() for n, p in model.named_parameters()
            }
            with t.no_grad():
                for n, p in model.named_parameters():
                    p.data.add_(state["noise"][n])

    def post_hook(module, inputs, output):
        if state["orig"] is not None:
            with t.no_grad():
                for n, p in model.named_parameters():
                    p.data.copy_(state["orig"][n])
            state["orig"] = None
            state["perturbed_out"] = output
        else:
            # clean pass: adapt sigma if we have a perturbed output
            if state["perturbed_out"] is not None and t.is_tensor(output):
                po = state["perturbed_out"]
                if t.is_tensor(po) and po.shape == output.shape:
                    spec.adapt(distance_func(output.detach(), po.detach()))
                state["perturbed_out"] = None

    h1 = model.register_forward_pre_hook(pre_hook)
    h2 = model.register_forward_hook(post_hook)

    def cancel():
        h1.remove()
        h2.remove()

    return cancel, spec
