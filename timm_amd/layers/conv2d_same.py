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


class PatchifyConv2d(nn.Conv2d):
    """Non-overlapping (stride == kernel, pad 0) conv lowered to
    reshape + hipBLASLt GEMM — MIOpen's bf16 NHWC path falls back to naive
    double-accum kernels for these shapes (ConvNeXt stem 4x4/s4 and 2x2/s2
    downsamples were 88% of remaining conv time, see gpurun_out/prof_cnx2).
    Identical params/state-dict to nn.Conv2d."""

    def forward(self, x):
        kh, kw = self.kernel_size
        B, C, H, W = x.shape
        if (
                x.is_cuda
                and self.stride[0] == kh and self.stride[1] == kw
                and self.padding[0] == 0 and self.padding[1] == 0
                and H % kh == 0 and W % kw == 0
        ):
            nh, nw = H // kh, W // kw
            patches = x.view(B, C, nh, kh, nw, kw).permute(0, 2, 4, 1, 3, 5).reshape(B, nh * nw, C * kh * kw)
            w = self.weight.reshape(self.out_channels, -1)
            y = torch.nn.functional.linear(patches, w, self.bias)
            # [B, nh, nw, O] -> NCHW view with channels-last strides
            return y.view(B, nh, nw, self.out_channels).permute(0, 3, 1, 2)
        return super().forward(x)


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
    ks = kernel_size if isinstance(kernel_size, int) else None
    if (
            ks is not None and padding == 0 and kwargs.get('groups', 1) == 1
            and kwargs.get('dilation', 1) == 1 and kwargs.get('stride', 1) == ks
    ):
        return PatchifyConv2d(in_chs, out_chs, kernel_size, padding=0, **kwargs)
    return nn.Conv2d(in_chs, out_chs, kernel_size, padding=padding, **kwargs)
