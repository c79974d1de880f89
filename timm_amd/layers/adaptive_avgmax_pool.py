"""Selectable global pooling.

Behavioral parity: /root/reference/timm/layers/adaptive_avgmax_pool.py
(pool_type strings, feat-mult, fast/NHWC constraints).  Redesigned around a
single mode-parameterized reduction instead of one class per mode: the four
Fast* class names survive as thin subclasses of ``FastGlobalPool`` so module
reprs and isinstance checks stay compatible.
"""
from typing import Tuple, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from .format import get_channel_dim, get_spatial_dim

_int_tuple_2_t = Union[int, Tuple[int, int]]

_CAT_MODES = ('catavgmax',)


def adaptive_pool_feat_mult(pool_type: str = 'avg') -> int:
    """Channel multiplier of the pooled output (2 for concat modes)."""
    return 2 if any(pool_type.endswith(m) for m in _CAT_MODES) else 1


def adaptive_avgmax_pool2d(x, output_size: _int_tuple_2_t = 1):
    return 0.5 * (F.adaptive_avg_pool2d(x, output_size) +
                  F.adaptive_max_pool2d(x, output_size))


def adaptive_catavgmax_pool2d(x, output_size: _int_tuple_2_t = 1):
    return torch.cat((
        F.adaptive_avg_pool2d(x, output_size),
        F.adaptive_max_pool2d(x, output_size),
    ), 1)


def select_adaptive_pool2d(x, pool_type: str = 'avg', output_size: _int_tuple_2_t = 1):
    """Functional NCHW adaptive pool selected by name."""
    fns = {
        'avg': F.adaptive_avg_pool2d,
        'max': F.adaptive_max_pool2d,
        'avgmax': adaptive_avgmax_pool2d,
        'catavgmax': adaptive_catavgmax_pool2d,
    }
    assert pool_type in fns, f'Invalid pool type: {pool_type}'
    return fns[pool_type](x, output_size)


class FastGlobalPool(nn.Module):
    """Global (output_size=1) pooling over the spatial dims of NCHW/NHWC maps.

    ``mode``: 'avg' | 'max' | 'avgmax' (mean of both) | 'catavgmax' (concat).
    """

    def __init__(self, mode: str = 'avg', flatten: bool = False, input_fmt: str = 'NCHW'):
        super().__init__()
        self.mode = mode
        self.flatten = flatten
        self.dim = get_spatial_dim(input_fmt)
        self.dim_cat = 1 if flatten else get_channel_dim(input_fmt)

    def forward(self, x):
        keep = not self.flatten
        if self.mode == 'avg':
            return x.mean(self.dim, keepdim=keep)
        if self.mode == 'max':
            return x.amax(self.dim, keepdim=keep)
        avg = x.mean(self.dim, keepdim=keep)
        mx = x.amax(self.dim, keepdim=keep)
        if self.mode == 'catavgmax':
            return torch.cat((avg, mx), self.dim_cat)
        return 0.5 * avg + 0.5 * mx


class FastAdaptiveAvgPool(FastGlobalPool):
    def __init__(self, flatten: bool = False, input_fmt: str = 'NCHW'):
        super().__init__('avg', flatten, input_fmt)


class FastAdaptiveMaxPool(FastGlobalPool):
    def __init__(self, flatten: bool = False, input_fmt: str = 'NCHW'):
        super().__init__('max', flatten, input_fmt)


class FastAdaptiveAvgMaxPool(FastGlobalPool):
    def __init__(self, flatten: bool = False, input_fmt: str = 'NCHW'):
        super().__init__('avgmax', flatten, input_fmt)


class FastAdaptiveCatAvgMaxPool(FastGlobalPool):
    def __init__(self, flatten: bool = False, input_fmt: str = 'NCHW'):
        super().__init__('catavgmax', flatten, input_fmt)


class AdaptiveAvgMaxPool2d(nn.Module):
    def __init__(self, output_size: _int_tuple_2_t = 1):
        super().__init__()
        self.output_size = output_size

    def forward(self, x):
        return adaptive_avgmax_pool2d(x, self.output_size)


class AdaptiveCatAvgMaxPool2d(nn.Module):
    def __init__(self, output_size: _int_tuple_2_t = 1):
        super().__init__()
        self.output_size = output_size

    def forward(self, x):
        return adaptive_catavgmax_pool2d(x, self.output_size)


class SelectAdaptivePool2d(nn.Module):
    """Global pooling layer selected by pool_type string.

    'fast*' variants (and any non-NCHW input) reduce with tensor ops and
    require output_size == 1; classic variants use torch adaptive pools.
    Empty pool_type passes through.
    """

    def __init__(
            self,
            output_size: _int_tuple_2_t = 1,
            pool_type: str = 'fast',
            flatten: bool = False,
            input_fmt: str = 'NCHW',
    ):
        super().__init__()
        assert input_fmt in ('NCHW', 'NHWC')
        self.pool_type = pool_type or ''
        key = self.pool_type.lower()

        if not key:
            self.pool = nn.Identity()
            self.flatten = nn.Flatten(1) if flatten else nn.Identity()
            return

        if key.startswith('fast') or input_fmt != 'NCHW':
            assert output_size == 1, \
                'Fast pooling and non NCHW input formats require output_size == 1.'
            mode = key[4:].lstrip('_') if key.startswith('fast') else key
            mode = mode or 'avg'
            assert mode in ('avg', 'max', 'avgmax', 'catavgmax'), \
                f'Invalid pool type: {pool_type}'
            self.pool = FastGlobalPool(mode, flatten, input_fmt=input_fmt)
            self.flatten = nn.Identity()
            return

        classic = {
            'avg': lambda: nn.AdaptiveAvgPool2d(output_size),
            'max': lambda: nn.AdaptiveMaxPool2d(output_size),
            'avgmax': lambda: AdaptiveAvgMaxPool2d(output_size),
            'catavgmax': lambda: AdaptiveCatAvgMaxPool2d(output_size),
        }
        assert key in classic, f'Invalid pool type: {pool_type}'
        self.pool = classic[key]()
        self.flatten = nn.Flatten(1) if flatten else nn.Identity()

    def is_identity(self):
        return not self.pool_type

    def forward(self, x):
        return self.flatten(self.pool(x))

    def feat_mult(self):
        return adaptive_pool_feat_mult(self.pool_type)

    def __repr__(self):
        return (f'{self.__class__.__name__}(pool_type={self.pool_type}'
                f', flatten={self.flatten})')
