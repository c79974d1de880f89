"""Activations + factory (parity with reference `timm/layers/activations.py`,
`create_act.py`).  Device-side these mostly appear fused into the GEMM
epilogue (`ops.bias_act`); the standalone modules are used elsewhere."""
from typing import Callable, Optional, Type

import torch
from torch import nn
from torch.nn import functional as F


def swish(x, inplace: bool = False):
    return x.mul_(x.sigmoid()) if inplace else x.mul(x.sigmoid())


class Swish(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()
        self.inplace = inplace

    def forward(self, x):
        return swish(x, self.inplace)


SiLU = nn.SiLU


def mish(x, inplace: bool = False):
    return x.mul(F.softplus(x).tanh())


class Mish(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()

    def forward(self, x):
        return mish(x)


def sigmoid(x, inplace: bool = False):
    return x.sigmoid_() if inplace else x.sigmoid()


class Sigmoid(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()
        self.inplace = inplace

    def forward(self, x):
        return x.sigmoid_() if self.inplace else x.sigmoid()


class Tanh(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()

    def forward(self, x):
        return x.tanh()


def hard_swish(x, inplace: bool = False):
    inner = F.relu6(x + 3.).div_(6.)
    return x.mul_(inner) if inplace else x.mul(inner)


class HardSwish(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()
        self.inplace = inplace

    def forward(self, x):
        return hard_swish(x, self.inplace)


def hard_sigmoid(x, inplace: bool = False):
    if inplace:
        return x.add_(3.).clamp_(0., 6.).div_(6.)
    return F.relu6(x + 3.) / 6.


class HardSigmoid(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()
        self.inplace = inplace

    def forward(self, x):
        return hard_sigmoid(x, self.inplace)


def hard_mish(x, inplace: bool = False):
    if inplace:
        return x.mul_(0.5 * (x + 2).clamp(min=0, max=2))
    return 0.5 * x * (x + 2).clamp(min=0, max=2)


class HardMish(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()
        self.inplace = inplace

    def forward(self, x):
        return hard_mish(x, self.inplace)


class PReLU(nn.PReLU):
    def __init__(self, num_parameters: int = 1, init: float = 0.25, inplace: bool = False):
        super().__init__(num_parameters=num_parameters, init=init)

    def forward(self, x):
        return F.prelu(x, self.weight)


def gelu(x, inplace: bool = False):
    return F.gelu(x)


class GELU(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()

    def forward(self, x):
        return F.gelu(x)


class GELUTanh(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()

    def forward(self, x):
        return F.gelu(x, approximate='tanh')


def quick_gelu(x, inplace: bool = False):
    return x * torch.sigmoid(1.702 * x)


class QuickGELU(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()

    def forward(self, x):
        return quick_gelu(x)
