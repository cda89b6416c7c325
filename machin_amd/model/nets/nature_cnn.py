"""Nature-CNN actor-critic stack (Atari) and MLP builders.

The reference ships only MLP/ResNet (machin/model/nets/); the Atari
CNN is the flagship bench model named by BASELINE.json (IMPALA Atari
CNN). Convs run through MIOpen, linears through hipBLASLt; channels
last memory format is used on ROCm for better MFMA-backed conv
kernels.
"""
from typing import Sequence

import torch as t
import torch.nn as nn

from .base import NeuralNetworkModule


class NatureCNN(NeuralNetworkModule):
    """The DQN-Nature convolutional torso: 32x8x8s4, 64x4x4s2,
    64x3x3s1, then a 512-unit linear; input [B, frames, 84, 84] scaled
    to [0, 1]."""

    def __init__(self, in_channels: int = 4, feature_dim: int = 512):
        super().__init__()
        self.conv = nn.Sequential(
            nn.Conv2d(in_channels, 32, 8, stride=4),
            nn.ReLU(inplace=True),
            nn.Conv2d(32, 64, 4, stride=2),
            nn.ReLU(inplace=True),
            nn.Conv2d(64, 64, 3, stride=1),
            nn.ReLU(inplace=True),
        )
        self.fc = nn.Sequential(nn.Flatten(), nn.Linear(64 * 7 * 7, feature_dim),
                                nn.ReLU(inplace=True))
        self.set_input_module(self.conv)
        self.set_output_module(self.fc)

    def forward(self, frames: t.Tensor) -> t.Tensor:
        return self.fc(self.conv(frames))


class ActorCriticCNN(NeuralNetworkModule):
    """Nature-CNN torso with policy-logit and value heads (IMPALA /
    A2C / PPO on Atari)."""

    def __init__(self, in_channels: int = 4, action_num: int = 6,
                 feature_dim: int = 512):
        super().__init__()
        self.torso = NatureCNN(in_channels, feature_dim)
        self.policy = nn.Linear(feature_dim, action_num)
        self.value = nn.Linear(feature_dim, 1)
        self.set_input_module(self.torso.conv)
        self.set_output_module(self.policy)

    def forward(self, frames: t.Tensor):
        feat = self.torso(frames)
        return self.policy(feat), self.value(feat)


def mlp(sizes: Sequence[int], activation=nn.ReLU, output_activation=None):
    """Plain MLP builder."""
    layers = []
    for i in range(len(sizes) - 1):
        layers.append(nn.Linear(sizes[i], sizes[i + 1]))
        if i < len(sizes) - 2:
            layers.append(activation())
        elif output_activation is not None:
            layers.append(output_activation())
    return nn.Sequential(*layers)


class FusedActorCriticCNN(NeuralNetworkModule):
    """ActorCriticCNN variant consuming RAW uint8 frames: the stem is
    machin_amd.ops.fused_conv.FusedAtariConv1 (fused u8 dequant +
    conv, hand-written gfx950 weight-gradient kernel); the rest of the
    stack is identical to ActorCriticCNN."""

    def __init__(self, action_num: int = 6, feature_dim: int = 512):
        super().__init__()
        from ...ops.fused_conv import FusedAtariConv1

        self.stem = FusedAtariConv1()
        self.conv_rest = nn.Sequential(
            nn.Conv2d(32, 64, 4, stride=2),
            nn.ReLU(inplace=True),
            nn.Conv2d(64, 64, 3, stride=1),
            nn.ReLU(inplace=True),
        )
        self.fc = nn.Sequential(
            nn.Flatten(), nn.Linear(64 * 7 * 7, feature_dim),
            nn.ReLU(inplace=True),
        )
        self.policy = nn.Linear(feature_dim, action_num)
        self.value = nn.Linear(feature_dim, 1)

    def forward(self, frames_u8: t.Tensor):
        x = t.relu(self.stem(frames_u8))
        with t.autocast(device_type="cuda", dtype=t.bfloat16):
            feat = self.fc(self.conv_rest(x))
            return self.policy(feat), self.value(feat)
