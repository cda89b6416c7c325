"""ResNet for RL vision stacks.

Parity target: reference ``machin/model/nets/resnet.py`` (:38-344):
configurable depth 18/34/50/101/152, basic and bottleneck blocks,
optional weight-norm instead of batch-norm (useful for RL where batch
statistics are non-stationary).

MI355X note: convolutions run through MIOpen; construct with
``.to(memory_format=torch.channels_last)`` for NHWC conv kernels.
"""
from typing import List, Type, Union

import torch as t
import torch.nn as nn

from .base import NeuralNetworkModule


def _norm2d(norm: str, planes: int):
    if norm == "batch":
        return nn.BatchNorm2d(planes)
    if norm == "none":
        return nn.Identity()
    raise ValueError(f"Unknown norm {norm!r}")


def _conv3x3(in_planes, out_planes, stride=1, weight_norm=False):
    conv = nn.Conv2d(in_planes, out_planes, 3, stride=stride, padding=1,
                     bias=False)
    return nn.utils.parametrizations.weight_norm(conv) if weight_norm else conv


def _conv1x1(in_planes, out_planes, stride=1, weight_norm=False):
    conv = nn.Conv2d(in_planes, out_planes, 1, stride=stride, bias=False)
    return nn.utils.parametrizations.weight_norm(conv) if weight_norm else conv


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_planes, planes, stride=1, norm="batch",
                 weight_norm=False):
        super().__init__()
        self.conv1 = _conv3x3(in_planes, planes, stride, weight_norm)
        self.bn1 = _norm2d(norm, planes)
        self.conv2 = _conv3x3(planes, planes, 1, weight_norm)
        self.bn2 = _norm2d(norm, planes)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != self.expansion * planes:
            self.shortcut = nn.Sequential(
                _conv1x1(in_planes, self.expansion * planes, stride,
                         weight_norm),
                _norm2d(norm, self.expansion * planes),
            )

    def forward(self, x):
        out = t.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        out = out + self.shortcut(x)
        return t.relu(out)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_planes, planes, stride=1, norm="batch",
                 weight_norm=False):
        super().__init__()
        self.conv1 = _conv1x1(in_planes, planes, 1, weight_norm)
        self.bn1 = _norm2d(norm, planes)
        self.conv2 = _conv3x3(planes, planes, stride, weight_norm)
        self.bn2 = _norm2d(norm, planes)
        self.conv3 = _conv1x1(planes, self.expansion * planes, 1, weight_norm)
        self.bn3 = _norm2d(norm, self.expansion * planes)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != self.expansion * planes:
            self.shortcut = nn.Sequential(
                _conv1x1(in_planes, self.expansion * planes, stride,
                         weight_norm),
                _norm2d(norm, self.expansion * planes),
            )

    def forward(self, x):
        out = t.relu(self.bn1(self.conv1(x)))
        out = t.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        out = out + self.shortcut(x)
        return t.relu(out)


_DEPTH_CONFIG = {
    18: (BasicBlock, [2, 2, 2, 2]),
    34: (BasicBlock, [3, 4, 6, 3]),
    50: (Bottleneck, [3, 4, 6, 3]),
    101: (Bottleneck, [3, 4, 23, 3]),
    152: (Bottleneck, [3, 8, 36, 3]),
}


class ResNet(NeuralNetworkModule):
    """ResNet-{18,34,50,101,152} with a configurable output head."""

    def __init__(
        self,
        in_planes: int,
        depth: int,
        out_planes: int,
        out_pool_size: Union[int, tuple] = 1,
        norm: str = "batch",
        weight_norm: bool = False,
        base_width: int = 64,
    ):
        super().__init__()
        if depth not in _DEPTH_CONFIG:
            raise ValueError(
                f"Depth must be one of {sorted(_DEPTH_CONFIG)}, got {depth}."
            )
        block, layers = _DEPTH_CONFIG[depth]
        self._in_planes = base_width
        self.conv1 = nn.Conv2d(in_planes, base_width, 7, stride=2,
                               padding=3, bias=False)
        self.bn1 = _norm2d(norm, base_width)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, base_width, layers[0], 1,
                                       norm, weight_norm)
        self.layer2 = self._make_layer(block, base_width * 2, layers[1], 2,
                                       norm, weight_norm)
        self.layer3 = self._make_layer(block, base_width * 4, layers[2], 2,
                                       norm, weight_norm)
        self.layer4 = self._make_layer(block, base_width * 8, layers[3], 2,
                                       norm, weight_norm)
        self.avgpool = nn.AdaptiveAvgPool2d(out_pool_size)
        pool = (
            out_pool_size
            if isinstance(out_pool_size, tuple)
            else (out_pool_size, out_pool_size)
        )
        self.fc = nn.Linear(
            base_width * 8 * block.expansion * pool[0] * pool[1], out_planes
        )
        self.set_input_module(self.conv1)
        self.set_output_module(self.fc)

    def _make_layer(self, block, planes, num_blocks, stride, norm,
                    weight_norm):
        strides = [stride] + [1] * (num_blocks - 1)
        layers = []
        for s in strides:
            layers.append(
                block(self._in_planes, planes, s, norm, weight_norm)
            )
            self._in_planes = planes * block.expansion
        return nn.Sequential(*layers)

    def forward(self, x):
        out = t.relu(self.bn1(self.conv1(x)))
        out = self.maxpool(out)
        out = self.layer4(self.layer3(self.layer2(self.layer1(out))))
        out = self.avgpool(out).flatten(1)
        return self.fc(out)
