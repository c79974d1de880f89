"""EvoNorm normalization+activation layers (reference `timm/layers/evo_norm.py`).

Paper: Evolving Normalization-Activation Layers — https://arxiv.org/abs/2004.02967
B-variants track running batch variance; S-variants use group std/rms.  Each
is a fused norm+nonlinearity replacing BN+act pairs.
"""
from typing import Optional, Sequence, Type, Union

import torch
import torch.nn as nn

from .create_act import create_act_layer

_V_SHAPE = (1, -1, 1, 1)


def instance_std(x: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    std = x.float().var(dim=(2, 3), unbiased=False, keepdim=True).add(eps).sqrt().to(x.dtype)
    return std.expand(x.shape)


def instance_rms(x: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    rms = x.float().square().mean(dim=(2, 3), keepdim=True).add(eps).sqrt().to(x.dtype)
    return rms.expand(x.shape)


def group_std(x: torch.Tensor, groups: int = 32, eps: float = 1e-5) -> torch.Tensor:
    B, C, H, W = x.shape
    assert C % groups == 0
    x_dtype = x.dtype
    xg = x.reshape(B, groups, C // groups, H, W)
    std = xg.float().var(dim=(2, 3, 4), unbiased=False, keepdim=True).add(eps).sqrt().to(x_dtype)
    return std.expand(xg.shape).reshape(B, C, H, W)


def group_rms(x: torch.Tensor, groups: int = 32, eps: float = 1e-5) -> torch.Tensor:
    B, C, H, W = x.shape
    assert C % groups == 0
    x_dtype = x.dtype
    xg = x.reshape(B, groups, C // groups, H, W)
    rms = xg.float().square().mean(dim=(2, 3, 4), keepdim=True).add(eps).sqrt_().to(x_dtype)
    return rms.expand(xg.shape).reshape(B, C, H, W)


class _EvoNormB(nn.Module):
    """Shared state for B-variants (running batch variance)."""

    def __init__(self, num_features: int, apply_act: bool = True, momentum: float = 0.1, eps: float = 1e-5):
        super().__init__()
        self.apply_act = apply_act
        self.momentum = momentum
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer('running_var', torch.ones(num_features))

    def reset_parameters(self):
        nn.init.ones_(self.weight)
        nn.init.zeros_(self.bias)

    def _batch_std(self, x: torch.Tensor) -> torch.Tensor:
        if self.training:
            var = x.float().var(dim=(0, 2, 3), unbiased=False)
            n = x.numel() / x.shape[1]
            self.running_var.copy_(
                self.running_var * (1 - self.momentum) +
                var.detach().to(self.running_var.dtype) * self.momentum * (n / (n - 1)))
        else:
            var = self.running_var
        return var.to(x.dtype).view(_V_SHAPE).add(self.eps).sqrt_()


class EvoNorm2dB0(_EvoNormB):
    def __init__(self, num_features, apply_act=True, momentum=0.1, eps=1e-3, **_):
        super().__init__(num_features, apply_act, momentum, eps)
        self.v = nn.Parameter(torch.ones(num_features)) if apply_act else None

    def reset_parameters(self):
        super().reset_parameters()
        if self.v is not None:
            nn.init.ones_(self.v)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        assert x.dim() == 4, 'expected 4D input'
        x_dtype = x.dtype
        if self.v is not None:
            left = self._batch_std(x).expand_as(x)
            v = self.v.to(x_dtype).view(_V_SHAPE)
            right = x * v + instance_std(x, self.eps)
            x = x / left.max(right)
        return x * self.weight.to(x_dtype).view(_V_SHAPE) + self.bias.to(x_dtype).view(_V_SHAPE)


class EvoNorm2dB1(_EvoNormB):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        assert x.dim() == 4, 'expected 4D input'
        x_dtype = x.dtype
        if self.apply_act:
            left = self._batch_std(x)
            right = (x + 1) * instance_rms(x, self.eps)
            x = x / left.max(right)
        return x * self.weight.view(_V_SHAPE).to(x_dtype) + self.bias.view(_V_SHAPE).to(x_dtype)


class EvoNorm2dB2(_EvoNormB):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        assert x.dim() == 4, 'expected 4D input'
        x_dtype = x.dtype
        if self.apply_act:
            left = self._batch_std(x)
            right = instance_rms(x, self.eps) - x
            x = x / left.max(right)
        return x * self.weight.view(_V_SHAPE).to(x_dtype) + self.bias.view(_V_SHAPE).to(x_dtype)


class EvoNorm2dS0(nn.Module):
    def __init__(self, num_features, groups=32, group_size=None, apply_act=True, eps=1e-5, **_):
        super().__init__()
        self.apply_act = apply_act
        if group_size:
            assert num_features % group_size == 0
            self.groups = num_features // group_size
        else:
            self.groups = groups
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.v = nn.Parameter(torch.ones(num_features)) if apply_act else None

    def reset_parameters(self):
        nn.init.ones_(self.weight)
        nn.init.zeros_(self.bias)
        if self.v is not None:
            nn.init.ones_(self.v)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        assert x.dim() == 4, 'expected 4D input'
        x_dtype = x.dtype
        if self.v is not None:
            v = self.v.view(_V_SHAPE).to(x_dtype)
            x = x * (x * v).sigmoid() / group_std(x, self.groups, self.eps)
        return x * self.weight.view(_V_SHAPE).to(x_dtype) + self.bias.view(_V_SHAPE).to(x_dtype)


class EvoNorm2dS0a(EvoNorm2dS0):
    def __init__(self, num_features, groups=32, group_size=None, apply_act=True, eps=1e-3, **_):
        super().__init__(num_features, groups=groups, group_size=group_size, apply_act=apply_act, eps=eps)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        assert x.dim() == 4, 'expected 4D input'
        x_dtype = x.dtype
        d = group_std(x, self.groups, self.eps)
        if self.v is not None:
            v = self.v.view(_V_SHAPE).to(x_dtype)
            x = x * (x * v).sigmoid()
        x = x / d
        return x * self.weight.view(_V_SHAPE).to(x_dtype) + self.bias.view(_V_SHAPE).to(x_dtype)


class EvoNorm2dS1(nn.Module):
    def __init__(
            self, num_features, groups=32, group_size=None, apply_act=True,
            act_layer: Optional[Type[nn.Module]] = None, eps=1e-5, **_):
        super().__init__()
        act_layer = act_layer or nn.SiLU
        self.apply_act = apply_act
        if act_layer is not None and apply_act:
            self.act = create_act_layer(act_layer)
        else:
            self.act = nn.Identity()
        if group_size:
            assert num_features % group_size == 0
            self.groups = num_features // group_size
        else:
            self.groups = groups
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))

    def reset_parameters(self):
        nn.init.ones_(self.weight)
        nn.init.zeros_(self.bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        assert x.dim() == 4, 'expected 4D input'
        x_dtype = x.dtype
        if self.apply_act:
            x = self.act(x) / group_std(x, self.groups, self.eps)
        return x * self.weight.view(_V_SHAPE).to(x_dtype) + self.bias.view(_V_SHAPE).to(x_dtype)


class EvoNorm2dS1a(EvoNorm2dS1):
    def __init__(
            self, num_features, groups=32, group_size=None, apply_act=True,
            act_layer=None, eps=1e-3, **_):
        super().__init__(
            num_features, groups=groups, group_size=group_size, apply_act=apply_act,
            act_layer=act_layer, eps=eps)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        assert x.dim() == 4, 'expected 4D input'
        x_dtype = x.dtype
        x = self.act(x) / group_std(x, self.groups, self.eps)
        return x * self.weight.view(_V_SHAPE).to(x_dtype) + self.bias.view(_V_SHAPE).to(x_dtype)


class EvoNorm2dS2(EvoNorm2dS1):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        assert x.dim() == 4, 'expected 4D input'
        x_dtype = x.dtype
        if self.apply_act:
            x = self.act(x) / group_rms(x, self.groups, self.eps)
        return x * self.weight.view(_V_SHAPE).to(x_dtype) + self.bias.view(_V_SHAPE).to(x_dtype)


class EvoNorm2dS2a(EvoNorm2dS2):
    def __init__(
            self, num_features, groups=32, group_size=None, apply_act=True,
            act_layer=None, eps=1e-3, **_):
        super().__init__(
            num_features, groups=groups, group_size=group_size, apply_act=apply_act,
            act_layer=act_layer, eps=eps)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        assert x.dim() == 4, 'expected 4D input'
        x_dtype = x.dtype
        x = self.act(x) / group_rms(x, self.groups, self.eps)
        return x * self.weight.view(_V_SHAPE).to(x_dtype) + self.bias.view(_V_SHAPE).to(x_dtype)
