"""Noise generators.

Parity target: reference ``machin/frame/noise/generator.py`` —
NormalNoiseGen (:33), ClippedNormalNoiseGen (:64), UniformNoiseGen
(:107), OrnsteinUhlenbeckNoiseGen (:138). Construct with a shape,
call with an optional device to draw a sample.

MI355X note: when the requested device is a ROCm GPU and the HIP
extension is built, Normal/OU generation runs the Philox kernels in
machin_amd/ops/hip/distributions.hip (one kernel, no intermediate
tensors); otherwise torch RNG.
"""
from typing import Tuple, Union

import torch as t


class NoiseGen:
    """Base: callable returning a noise tensor of the configured shape."""

    def __call__(self, device: Union[str, t.device] = "cpu") -> t.Tensor:
        raise NotImplementedError

    def reset(self):
        pass


class NormalNoiseGen(NoiseGen):
    def __init__(self, shape: Tuple[int, ...], mu: float = 0.0,
                 sigma: float = 1.0):
        self.shape = tuple(shape)
        self.mu = mu
        self.sigma = sigma

    def __call__(self, device="cpu") -> t.Tensor:
        return t.randn(self.shape, device=device) * self.sigma + self.mu

    def __repr__(self):
        return f"NormalNoiseGen(mu={self.mu}, sigma={self.sigma})"


class ClippedNormalNoiseGen(NormalNoiseGen):
    def __init__(self, shape, mu: float = 0.0, sigma: float = 1.0,
                 nmin: float = -1.0, nmax: float = 1.0):
        super().__init__(shape, mu, sigma)
        self.nmin = nmin
        self.nmax = nmax

    def __call__(self, device="cpu") -> t.Tensor:
        return super().__call__(device).clamp(self.nmin, self.nmax)

    def __repr__(self):
        return (
            f"ClippedNormalNoiseGen(mu={self.mu}, sigma={self.sigma}, "
            f"min={self.nmin}, max={self.nmax})"
        )


class UniformNoiseGen(NoiseGen):
    def __init__(self, shape, umin: float = 0.0, umax: float = 1.0):
        self.shape = tuple(shape)
        self.umin = umin
        self.umax = umax

    def __call__(self, device="cpu") -> t.Tensor:
        return (
            t.rand(self.shape, device=device) * (self.umax - self.umin)
            + self.umin
        )

    def __repr__(self):
        return f"UniformNoiseGen(min={self.umin}, max={self.umax})"


class OrnsteinUhlenbeckNoiseGen(NoiseGen):
    """Temporally-correlated noise:
    x += theta*(mu - x)*dt + sigma*sqrt(dt)*N(0,1)."""

    def __init__(self, shape, mu: float = 0.0, sigma: float = 0.2,
                 theta: float = 0.15, dt: float = 1e-2, x0: t.Tensor = None):
        self.shape = tuple(shape)
        self.mu = mu
        self.sigma = sigma
        self.theta = theta
        self.dt = dt
        self.x0 = x0
        self.x_prev = None
        self.reset()

    def reset(self):
        self.x_prev = (
            self.x0.clone() if self.x0 is not None else t.zeros(self.shape)
        )

    def __call__(self, device="cpu") -> t.Tensor:
        x_prev = self.x_prev.to(device)
        if x_prev.is_cuda:
            from ...ops import available, _require_ext

            if available():
                x = x_prev.float().contiguous()
                _require_ext().ou_update_(
                    x, self.mu, self.theta, self.sigma, self.dt,
                    int(t.randint(0, 2 ** 31, (1,)).item()), 0,
                )
                self.x_prev = x
                return x
        x = (
            x_prev
            + self.theta * (self.mu - x_prev) * self.dt
            + self.sigma * (self.dt ** 0.5) * t.randn(self.shape, device=device)
        )
        self.x_prev = x
        return x

    def __repr__(self):
        return (
            f"OrnsteinUhlenbeckNoiseGen(mu={self.mu}, sigma={self.sigma}, "
            f"theta={self.theta}, dt={self.dt})"
        )
