"""Activation modules with memory-efficient analytic backward.

Capability parity with reference models/activations.py (= the timm copy):
the JIT-scripted Swish/Mish with hand-written backward become module
wrappers over the HIP activation kernels (csrc/activations.hip), which use
the same recompute-sigmoid trick.
"""

import torch
from torch import nn

from .. import ops


class Swish(nn.Module):
    def __init__(self, inplace=False):
        super().__init__()

    def forward(self, x):
        return ops.swish(x)


MemoryEfficientSwish = Swish  # reference exposes both names


class Mish(nn.Module):
    def __init__(self, inplace=False):
        super().__init__()

    def forward(self, x):
        return ops.mish(x)


MemoryEfficientMish = Mish


def swish(x, inplace=False):
    return ops.swish(x)


def mish(x, inplace=False):
    return ops.mish(x)


def hard_swish(x, inplace=False):
    return ops.hard_swish(x)


def hard_sigmoid(x, inplace=False):
    return ops.hard_sigmoid(x)


class HardSwish(nn.Module):
    def __init__(self, inplace=False):
        super().__init__()

    def forward(self, x):
        return ops.hard_swish(x)


class HardSigmoid(nn.Module):
    def __init__(self, inplace=False):
        super().__init__()

    def forward(self, x):
        return ops.hard_sigmoid(x)


class Sigmoid(nn.Module):
    def __init__(self, inplace=False):
        super().__init__()

    def forward(self, x):
        return ops.sigmoid(x)


class Tanh(nn.Module):
    def __init__(self, inplace=False):
        super().__init__()

    def forward(self, x):
        return torch.tanh(x)
