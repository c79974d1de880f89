"""Non-local attention + grouped bilinear attention transform (BAT).

Capability parity with reference `timm/layers/non_local_attn.py` — the
spatial non-local block (`NonLocalAttn`, :19) and the BAT block
(`BilinearAttnTransform` :87 / `BatNonLocalAttn` :148) used by the *nl / *bat
ResNet variants.  The non-local path is three 1x1 convs + two batched GEMMs,
which lower straight onto hipBLASLt on MI355X.
"""
from typing import Optional, Type

import torch
from torch import nn
from torch.nn import functional as F

from .conv_bn_act import ConvNormAct
from .helpers import make_divisible
from .trace_utils import _assert

__all__ = ['NonLocalAttn', 'BatNonLocalAttn']


class NonLocalAttn(nn.Module):
    """Classic spatial non-local block (video-nonlocal-net style)."""

    def __init__(
            self,
            in_channels: int,
            use_scale: bool = True,
            rd_ratio: float = 1 / 8,
            rd_channels: Optional[int] = None,
            rd_divisor: int = 8,
            **_,
    ):
        super().__init__()
        if rd_channels is None:
            rd_channels = make_divisible(in_channels * rd_ratio, divisor=rd_divisor)
        self.scale = in_channels ** -0.5 if use_scale else 1.0
        self.t = nn.Conv2d(in_channels, rd_channels, 1, bias=True)
        self.p = nn.Conv2d(in_channels, rd_channels, 1, bias=True)
        self.g = nn.Conv2d(in_channels, rd_channels, 1, bias=True)
        self.z = nn.Conv2d(rd_channels, in_channels, 1, bias=True)
        self.norm = nn.BatchNorm2d(in_channels)
        self.reset_parameters()

    def reset_parameters(self):
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode='fan_out', nonlinearity='relu')
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, (nn.BatchNorm2d, nn.GroupNorm)):
                # zero-init the output norm -> block starts as identity
                nn.init.zeros_(m.weight)
                nn.init.zeros_(m.bias)

    def forward(self, x):
        shortcut = x
        t = self.t(x)
        p = self.p(x)
        g = self.g(x)

        B, C, H, W = t.shape
        t = t.flatten(2).transpose(1, 2)       # B, HW, C
        p = p.flatten(2)                       # B, C, HW
        g = g.flatten(2).transpose(1, 2)       # B, HW, C

        att = torch.bmm(t, p) * self.scale
        att = F.softmax(att, dim=2)
        y = torch.bmm(att, g).transpose(1, 2).reshape(B, C, H, W)
        y = self.z(y)
        return self.norm(y) + shortcut


class BilinearAttnTransform(nn.Module):
    """Grouped bilinear attention transform y = p @ x @ q."""

    def __init__(
            self,
            in_channels: int,
            block_size: int,
            groups: int,
            act_layer: Type[nn.Module] = nn.ReLU,
            norm_layer: Type[nn.Module] = nn.BatchNorm2d,
    ):
        super().__init__()
        self.conv1 = ConvNormAct(in_channels, groups, 1, act_layer=act_layer, norm_layer=norm_layer)
        self.conv_p = nn.Conv2d(groups, block_size * block_size * groups, kernel_size=(block_size, 1))
        self.conv_q = nn.Conv2d(groups, block_size * block_size * groups, kernel_size=(1, block_size))
        self.conv2 = ConvNormAct(in_channels, in_channels, 1, act_layer=act_layer, norm_layer=norm_layer)
        self.block_size = block_size
        self.groups = groups
        self.in_channels = in_channels

    def resize_mat(self, x: torch.Tensor, t: int) -> torch.Tensor:
        """Expand a (bs x bs) attention matrix to (bs*t x bs*t) block-diagonal form."""
        B, C, bs, bs1 = x.shape
        _assert(bs == bs1, '')
        if t <= 1:
            return x
        x = x.view(B * C, -1, 1, 1) * torch.eye(t, t, dtype=x.dtype, device=x.device)
        x = x.view(B * C, bs, bs, t, t)
        x = torch.cat(torch.split(x, 1, dim=1), dim=3)
        x = torch.cat(torch.split(x, 1, dim=2), dim=4)
        return x.view(B, C, bs * t, bs * t)

    def forward(self, x):
        _assert(x.shape[-1] % self.block_size == 0, '')
        _assert(x.shape[-2] % self.block_size == 0, '')
        B, C, H, W = x.shape
        bs, g = self.block_size, self.groups
        out = self.conv1(x)
        rp = F.adaptive_max_pool2d(out, (bs, 1))
        cp = F.adaptive_max_pool2d(out, (1, bs))
        p = self.conv_p(rp).view(B, g, bs, bs).sigmoid()
        q = self.conv_q(cp).view(B, g, bs, bs).sigmoid()
        # row/col-stochastic normalization
        p = p / p.sum(dim=3, keepdim=True)
        q = q / q.sum(dim=2, keepdim=True)
        p = p.view(B, g, 1, bs, bs).expand(B, g, C // g, bs, bs).reshape(B, C, bs, bs)
        q = q.view(B, g, 1, bs, bs).expand(B, g, C // g, bs, bs).reshape(B, C, bs, bs)
        p = self.resize_mat(p, H // bs)
        q = self.resize_mat(q, W // bs)
        y = p.matmul(x).matmul(q)
        return self.conv2(y)


class BatNonLocalAttn(nn.Module):
    """BAT non-local block: reduce → bilinear transform → expand + residual."""

    def __init__(
            self,
            in_channels: int,
            block_size: int = 7,
            groups: int = 2,
            rd_ratio: float = 0.25,
            rd_channels: Optional[int] = None,
            rd_divisor: int = 8,
            drop_rate: float = 0.2,
            act_layer: Type[nn.Module] = nn.ReLU,
            norm_layer: Type[nn.Module] = nn.BatchNorm2d,
            **_,
    ):
        super().__init__()
        if rd_channels is None:
            rd_channels = make_divisible(in_channels * rd_ratio, divisor=rd_divisor)
        self.conv1 = ConvNormAct(in_channels, rd_channels, 1, act_layer=act_layer, norm_layer=norm_layer)
        self.ba = BilinearAttnTransform(rd_channels, block_size, groups, act_layer=act_layer, norm_layer=norm_layer)
        self.conv2 = ConvNormAct(rd_channels, in_channels, 1, act_layer=act_layer, norm_layer=norm_layer)
        self.dropout = nn.Dropout2d(p=drop_rate)

    def forward(self, x):
        y = self.conv1(x)
        y = self.ba(y)
        y = self.conv2(y)
        y = self.dropout(y)
        return y + x
