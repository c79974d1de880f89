"""Coordinate attention and strip-pooling variants (reference `timm/layers/coord_attn.py`).

`CoordAttn` (CVPR'21 coordinate attention: joint H/W strip encoding),
`SimpleCoordAttn` (linear, additive recombination), `EfficientLocalAttn`
(1D depthwise convs + GroupNorm, arXiv:2403.01123) and `StripAttn`.
"""
from typing import Optional, Type, Union

import torch
from torch import nn

from .create_act import create_act_layer
from .helpers import make_divisible
from .norm import GroupNorm1

__all__ = ['CoordAttn', 'SimpleCoordAttn', 'EfficientLocalAttn', 'StripAttn']


class CoordAttn(nn.Module):
    """Coordinate attention: H and W strips share a bottleneck conv, then
    separate per-axis gates multiply the input."""

    def __init__(
            self,
            channels: int,
            rd_ratio: float = 1. / 16,
            rd_channels: Optional[int] = None,
            rd_divisor: int = 8,
            se_factor: float = 2 / 3,
            bias: bool = False,
            act_layer: Type[nn.Module] = nn.Hardswish,
            norm_layer: Optional[Type[nn.Module]] = nn.BatchNorm2d,
            gate_layer: Union[str, Type[nn.Module]] = 'sigmoid',
            has_skip: bool = False,
    ):
        super().__init__()
        self.has_skip = has_skip
        if not rd_channels:
            rd_channels = make_divisible(channels * rd_ratio * se_factor, rd_divisor, round_limit=0.)

        self.conv1 = nn.Conv2d(channels, rd_channels, kernel_size=1, bias=bias)
        self.bn1 = norm_layer(rd_channels) if norm_layer is not None else nn.Identity()
        self.act = act_layer()
        self.conv_h = nn.Conv2d(rd_channels, channels, kernel_size=1, bias=bias)
        self.conv_w = nn.Conv2d(rd_channels, channels, kernel_size=1, bias=bias)
        self.gate = create_act_layer(gate_layer)

    def forward(self, x):
        identity = x
        N, C, H, W = x.shape

        x_h = x.mean(3, keepdim=True)                 # (N, C, H, 1)
        x_w = x.mean(2, keepdim=True).transpose(-1, -2)  # (N, C, W, 1)
        y = torch.cat([x_h, x_w], dim=2)
        y = self.act(self.bn1(self.conv1(y)))
        x_h, x_w = torch.split(y, [H, W], dim=2)
        x_w = x_w.transpose(-1, -2)

        a_h = self.gate(self.conv_h(x_h))
        a_w = self.gate(self.conv_w(x_w))

        out = identity * a_w * a_h
        if self.has_skip:
            out = out + identity
        return out


class SimpleCoordAttn(nn.Module):
    """Simplified coordinate attention: linear bottleneck, additive gate."""

    def __init__(
            self,
            channels: int,
            rd_ratio: float = 0.25,
            rd_channels: Optional[int] = None,
            rd_divisor: int = 8,
            se_factor: float = 2 / 3,
            bias: bool = True,
            act_layer: Type[nn.Module] = nn.SiLU,
            gate_layer: Union[str, Type[nn.Module]] = 'sigmoid',
            has_skip: bool = False,
    ):
        super().__init__()
        self.has_skip = has_skip
        if not rd_channels:
            rd_channels = make_divisible(channels * rd_ratio * se_factor, rd_divisor, round_limit=0.)

        self.fc1 = nn.Linear(channels, rd_channels, bias=bias)
        self.act = act_layer()
        self.fc_h = nn.Linear(rd_channels, channels, bias=bias)
        self.fc_w = nn.Linear(rd_channels, channels, bias=bias)
        self.gate = create_act_layer(gate_layer)

    def forward(self, x):
        identity = x

        x_h = x.mean(dim=3)   # (N, C, H)
        x_w = x.mean(dim=2)   # (N, C, W)

        x_h = self.act(self.fc1(x_h.transpose(1, 2)))  # (N, H, rd)
        x_w = self.act(self.fc1(x_w.transpose(1, 2)))  # (N, W, rd)

        a_h = self.fc_h(x_h).transpose(1, 2).unsqueeze(-1)  # (N, C, H, 1)
        a_w = self.fc_w(x_w).transpose(1, 2).unsqueeze(-2)  # (N, C, 1, W)

        out = identity * self.gate(a_h + a_w)
        if self.has_skip:
            out = out + identity
        return out


class EfficientLocalAttn(nn.Module):
    """Efficient local attention: per-axis 1D depthwise convs + GroupNorm."""

    def __init__(
            self,
            channels: int,
            kernel_size: int = 7,
            bias: bool = False,
            act_layer: Type[nn.Module] = nn.SiLU,
            gate_layer: Union[str, Type[nn.Module]] = 'sigmoid',
            norm_layer: Optional[Type[nn.Module]] = GroupNorm1,
            has_skip: bool = False,
    ):
        super().__init__()
        self.has_skip = has_skip

        self.conv_h = nn.Conv2d(
            channels, channels, kernel_size=(kernel_size, 1), stride=1,
            padding=(kernel_size // 2, 0), groups=channels, bias=bias)
        self.conv_w = nn.Conv2d(
            channels, channels, kernel_size=(1, kernel_size), stride=1,
            padding=(0, kernel_size // 2), groups=channels, bias=bias)
        if norm_layer is not None:
            self.norm_h = norm_layer(channels)
            self.norm_w = norm_layer(channels)
        else:
            self.norm_h = nn.Identity()
            self.norm_w = nn.Identity()
        self.act = act_layer()
        self.gate = create_act_layer(gate_layer)

    def forward(self, x):
        identity = x

        x_h = x.mean(dim=3, keepdim=True)
        x_w = x.mean(dim=2, keepdim=True)

        x_h = self.act(self.norm_h(self.conv_h(x_h)))
        x_w = self.act(self.norm_w(self.conv_w(x_w)))

        out = identity * self.gate(x_h) * self.gate(x_w)
        if self.has_skip:
            out = out + identity
        return out


class StripAttn(nn.Module):
    """Minimal strip attention: gate on summed H/W strip features."""

    def __init__(
            self,
            channels: int,
            use_conv: bool = True,
            kernel_size: int = 3,
            bias: bool = False,
            gate_layer: Union[str, Type[nn.Module]] = 'sigmoid',
            has_skip: bool = False,
            **_,
    ):
        super().__init__()
        self.has_skip = has_skip
        self.use_conv = use_conv

        if use_conv:
            self.conv_h = nn.Conv2d(
                channels, channels, kernel_size=(kernel_size, 1), stride=1,
                padding=(kernel_size // 2, 0), groups=channels, bias=bias)
            self.conv_w = nn.Conv2d(
                channels, channels, kernel_size=(1, kernel_size), stride=1,
                padding=(0, kernel_size // 2), groups=channels, bias=bias)
        else:
            self.conv_h = nn.Identity()
            self.conv_w = nn.Identity()

        self.gate = create_act_layer(gate_layer)

    def forward(self, x):
        identity = x

        x_h = self.conv_h(x.mean(dim=3, keepdim=True))
        x_w = self.conv_w(x.mean(dim=2, keepdim=True))

        out = identity * self.gate(x_h + x_w)
        if self.has_skip:
            out = out + identity
        return out
