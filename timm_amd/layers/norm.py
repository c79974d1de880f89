"""Norm layers (parity with reference `timm/layers/norm.py`, 575 LoC).

All last-dim norms route through `timm_amd.ops` → fused one-pass HIP kernels
on device (fp32 accumulation), replacing the reference's
`F.layer_norm`/`fast_norm` path (`timm/layers/fast_norm.py:119-160`).
2d variants operate on NCHW via permute (LayerNorm2d) with the same kernel.
"""
import numbers
from typing import Optional, Tuple, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops


class LayerNorm(nn.LayerNorm):
    """LayerNorm w/ fused HIP path (reference `norm.py:70`)."""

    def __init__(self, num_channels, eps=1e-6, affine=True, bias=True, **kwargs):
        super().__init__(num_channels, eps=eps, elementwise_affine=affine, bias=bias, **kwargs)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.layer_norm(x, self.normalized_shape, self.weight, self.bias, self.eps)


class LayerNormFp32(LayerNorm):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        dt = x.dtype
        w = self.weight.float() if self.weight is not None else None
        b = self.bias.float() if self.bias is not None else None
        return F.layer_norm(x.float(), self.normalized_shape, w, b, self.eps).to(dt)


class LayerNorm2d(nn.LayerNorm):
    """LayerNorm for NCHW tensors, normalizing over C (reference `norm.py:113`)."""

    def __init__(self, num_channels, eps=1e-6, affine=True, **kwargs):
        super().__init__(num_channels, eps=eps, elementwise_affine=affine, **kwargs)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x.permute(0, 2, 3, 1)
        x = ops.layer_norm(x, self.normalized_shape, self.weight, self.bias, self.eps)
        return x.permute(0, 3, 1, 2)


class RmsNorm(nn.Module):
    """RMSNorm w/ fused HIP path (reference `norm.py:202`)."""
    __constants__ = ['normalized_shape', 'eps', 'elementwise_affine']

    def __init__(self, channels, eps=1e-6, affine=True, device=None, dtype=None):
        factory_kwargs = {'device': device, 'dtype': dtype}
        super().__init__()
        normalized_shape = channels
        if isinstance(normalized_shape, numbers.Integral):
            normalized_shape = (normalized_shape,)
        self.normalized_shape = tuple(normalized_shape)
        self.eps = eps
        self.elementwise_affine = affine
        if self.elementwise_affine:
            self.weight = nn.Parameter(torch.empty(self.normalized_shape, **factory_kwargs))
        else:
            self.register_parameter('weight', None)
        self.reset_parameters()

    def reset_parameters(self):
        if self.elementwise_affine:
            nn.init.ones_(self.weight)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.rms_norm(x, self.normalized_shape, self.weight, self.eps)


class RmsNormFp32(RmsNorm):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        dt = x.dtype
        w = self.weight.float() if self.weight is not None else None
        return ops.rms_norm(x.float(), self.normalized_shape, w, self.eps).to(dt)


class RmsNorm2d(RmsNorm):
    """RMSNorm for NCHW over C (reference `norm.py:294`)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x.permute(0, 2, 3, 1)
        x = ops.rms_norm(x, self.normalized_shape, self.weight, self.eps)
        return x.permute(0, 3, 1, 2)


class SimpleNorm(nn.Module):
    """RMSNorm without centering or affine-bias; fp32 norm (reference `norm.py:394`)."""

    def __init__(self, channels, eps=1e-6, affine=True, device=None, dtype=None):
        factory_kwargs = {'device': device, 'dtype': dtype}
        super().__init__()
        normalized_shape = channels
        if isinstance(normalized_shape, numbers.Integral):
            normalized_shape = (normalized_shape,)
        self.normalized_shape = tuple(normalized_shape)
        self.eps = eps
        self.elementwise_affine = affine
        if affine:
            self.weight = nn.Parameter(torch.empty(self.normalized_shape, **factory_kwargs))
        else:
            self.register_parameter('weight', None)
        self.reset_parameters()

    def reset_parameters(self):
        if self.elementwise_affine:
            nn.init.ones_(self.weight)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.rms_norm(x, self.normalized_shape, self.weight, self.eps)


class SimpleNorm2d(SimpleNorm):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x.permute(0, 2, 3, 1)
        x = ops.rms_norm(x, self.normalized_shape, self.weight, self.eps)
        return x.permute(0, 3, 1, 2)


class GroupNorm(nn.GroupNorm):
    def __init__(self, num_channels, num_groups=32, eps=1e-5, affine=True, **kwargs):
        super().__init__(num_groups, num_channels, eps=eps, affine=affine, **kwargs)

    def forward(self, x):
        return F.group_norm(x, self.num_groups, self.weight, self.bias, self.eps)


class GroupNorm1(nn.GroupNorm):
    """Group normalization with 1 group == LayerNorm over all non-batch dims for NCHW conv nets."""

    def __init__(self, num_channels, **kwargs):
        super().__init__(1, num_channels, **kwargs)

    def forward(self, x):
        return F.group_norm(x, self.num_groups, self.weight, self.bias, self.eps)
