"""Conv2d w/ TF 'SAME' padding (reference `timm/layers/conv2d_same.py`)."""
from typing import Optional, Tuple, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from .padding import pad_same, pad_same_arg, is_static_pad


def conv2d_same(
        x,
        weight: torch.Tensor,
        bias: Optional[torch.Tensor] = None,
        stride: Tuple[int, int] = (1, 1),
        padding: Tuple[int, int] = (0, 0),
        dilation: Tuple[int, int] = (1, 1),
        groups: int = 1,
):
    x = pad_same(x, weight.shape[-2:], stride, dilation)
    return F.conv2d(x, weight, bias, stride, (0, 0), dilation, groups)


class Conv2dSame(nn.Conv2d):
    """Tensorflow like 'SAME' convolution wrapper for 2D convolutions."""

    def __init__(self, in_channels, out_channels, kernel_size, stride=1, padding=0, dilation=1, groups=1, bias=True):
        super().__init__(in_channels, out_channels, kernel_size, stride, 0, dilation, groups, bias)

    def forward(self, x):
        return conv2d_same(x, self.weight, self.bias, self.stride, self.padding, self.dilation, self.groups)


class Conv2dSameExport(nn.Conv2d):
    """ONNX-export-friendly version w/ fixed-size pad computed at first call."""

    def __init__(self, in_channels, out_channels, kernel_size, stride=1, padding=0, dilation=1, groups=1, bias=True):
        super().__init__(in_channels, out_channels, kernel_size, stride, 0, dilation, groups, bias)
        self.pad = None
        self.pad_input_size = (0, 0)

    def forward(self, x):
        input_size = x.size()[-2:]
        if self.pad is None:
            pad_arg = pad_same_arg(input_size, self.weight.size()[-2:], self.stride, self.dilation)
            self.pad = nn.ZeroPad2d(pad_arg)
            self.pad_input_size = input_size
        x = self.pad(x)
        return F.conv2d(x, self.weight, self.bias, self.stride, self.padding, self.dilation, self.groups)


class DepthwiseConv2d(nn.Conv2d):
    """Depthwise nn.Conv2d routed to the gfx950 NHWC HIP kernels when the
    input is channels-last bf16 on device; identical params/state-dict to
    nn.Conv2d."""

    def forward(self, x):
        from .. import ops
        if (
                x.is_cuda
                and x.dtype == torch.bfloat16
                and self.dilation[0] == 1 and self.dilation[1] == 1
                and self.kernel_size[0] == self.kernel_size[1]
                and self.stride[0] == self.stride[1]
                and self.padding[0] == self.padding[1]
                and x.is_contiguous(memory_format=torch.channels_last)
        ):
            return ops.depthwise_conv2d(
                x, self.weight, self.bias,
                stride=self.stride[0], padding=self.padding[0], dilation=1)
        return super().forward(x)


def create_conv2d_pad(in_chs, out_chs, kernel_size, **kwargs):
    from .padding import get_padding_value
    padding = kwargs.pop('padding', '')
    kwargs.setdefault('bias', False)
    padding, is_dynamic = get_padding_value(padding, kernel_size, **kwargs)
    if is_dynamic:
        return Conv2dSame(in_chs, out_chs, kernel_size, **kwargs)
    if kwargs.get('groups', 1) == in_chs and in_chs == out_chs and kwargs.get('dilation', 1) == 1:
        return DepthwiseConv2d(in_chs, out_chs, kernel_size, padding=padding, **kwargs)
    return nn.Conv2d(in_chs, out_chs, kernel_size, padding=padding, **kwargs)
