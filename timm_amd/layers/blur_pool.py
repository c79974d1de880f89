"""Anti-aliased downsampling (BlurPool, "Making Convolutions Shift-Invariant
Again") — reference `timm/layers/blur_pool.py` (`BlurPool2d` :20,
`create_aa` :104).

Implemented as a depthwise conv with a fixed binomial filter; runs through
`F.conv2d` with groups=C (MIOpen handles the fixed-weight depthwise fine, and
this op is rare/cold — stems and stage transitions only).
"""
from functools import partial
from typing import Callable, Optional, Type, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from .padding import get_padding


class BlurPool2d(nn.Module):
    """Blur + subsample with a binomial (Pascal row) low-pass filter."""

    def __init__(
            self,
            channels: Optional[int] = None,
            filt_size: int = 3,
            stride: int = 2,
            pad_mode: str = 'reflect',
    ) -> None:
        super().__init__()
        assert filt_size > 1
        self.channels = channels
        self.filt_size = filt_size
        self.stride = stride
        self.pad_mode = pad_mode
        self.padding = [get_padding(filt_size, stride, dilation=1)] * 4

        # binomial coefficients == row of Pascal's triangle
        coeffs = torch.tensor([float(_binom(filt_size - 1, k)) for k in range(filt_size)])
        blur_filter = (coeffs[:, None] * coeffs[None, :])
        blur_filter = blur_filter / blur_filter.sum()
        blur_filter = blur_filter[None, None, :, :]
        if channels is not None:
            blur_filter = blur_filter.repeat(self.channels, 1, 1, 1)
        self.register_buffer('filt', blur_filter, persistent=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = F.pad(x, self.padding, mode=self.pad_mode)
        if self.channels is None:
            channels = x.shape[1]
            weight = self.filt.expand(channels, 1, self.filt_size, self.filt_size)
        else:
            channels = self.channels
            weight = self.filt
        return F.conv2d(x, weight.to(dtype=x.dtype), stride=self.stride, groups=channels)


def _binom(n: int, k: int) -> int:
    import math
    return math.comb(n, k)


def create_aa(
        aa_layer: Union[str, Type[nn.Module], Callable, None],
        channels: Optional[int] = None,
        stride: int = 2,
        enable: bool = True,
        noop: Optional[Type[nn.Module]] = nn.Identity,
) -> Optional[nn.Module]:
    """Anti-aliasing layer factory (reference `blur_pool.py:104`)."""
    if not aa_layer or not enable:
        return noop() if noop is not None else None

    if isinstance(aa_layer, str):
        key = aa_layer.lower().replace('_', '').replace('-', '')
        if key in ('avg', 'avgpool'):
            aa_layer = nn.AvgPool2d
        elif key in ('blur', 'blurpool'):
            aa_layer = BlurPool2d
        elif key == 'blurpc':
            aa_layer = partial(BlurPool2d, pad_mode='constant')
        else:
            raise AssertionError(f'Unknown anti-aliasing layer ({aa_layer}).')

    try:
        return aa_layer(channels=channels, stride=stride)
    except TypeError:
        return aa_layer(stride)
