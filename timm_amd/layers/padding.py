"""Conv padding arithmetic, including TF-style dynamic 'SAME'
(reference `timm/layers/padding.py`).

Symmetric PyTorch padding is resolved statically whenever the
(kernel, stride, dilation) combination allows it; only genuinely asymmetric
'SAME' cases fall back to a runtime F.pad (extra kernel + memory traffic on
the GPU, so the static path is strongly preferred).
"""
import math
from typing import List, Tuple, Union

import torch
import torch.nn.functional as F


def _pair(v):
    return v if isinstance(v, (tuple, list)) else (v, v)


def get_padding(kernel_size: int, stride: int = 1, dilation: int = 1, **_) -> Union[int, List[int]]:
    """Symmetric padding that keeps output size == ceil(input / stride)."""
    if any(isinstance(v, (tuple, list)) for v in (kernel_size, stride, dilation)):
        return [
            get_padding(k, s, d)
            for k, s, d in zip(_pair(kernel_size), _pair(stride), _pair(dilation))
        ]
    return ((stride - 1) + dilation * (kernel_size - 1)) // 2


def get_same_padding(x: int, kernel_size: int, stride: int, dilation: int):
    """Total pad needed along one dim for TF 'SAME' output size."""
    if isinstance(x, torch.Tensor):
        return torch.clamp(-x % stride + (kernel_size - 1) * dilation + 1 - stride, min=0)
    return max((math.ceil(x / stride) - 1) * stride + (kernel_size - 1) * dilation + 1 - x, 0)


def is_static_pad(kernel_size: int, stride: int = 1, dilation: int = 1, **_):
    """True if 'SAME' padding is input-size independent (resolvable at build time)."""
    if any(isinstance(v, (tuple, list)) for v in (kernel_size, stride, dilation)):
        return all(
            is_static_pad(k, s, d)
            for k, s, d in zip(_pair(kernel_size), _pair(stride), _pair(dilation))
        )
    return stride == 1 and (dilation * (kernel_size - 1)) % 2 == 0


def pad_same_arg(
        input_size: List[int],
        kernel_size: List[int],
        stride: List[int],
        dilation: List[int] = (1, 1),
) -> List[int]:
    """F.pad argument (left, right, top, bottom) for 'SAME' given a known input size."""
    pad_h = get_same_padding(input_size[0], kernel_size[0], stride[0], dilation[0])
    pad_w = get_same_padding(input_size[1], kernel_size[1], stride[1], dilation[1])
    return [pad_w // 2, pad_w - pad_w // 2, pad_h // 2, pad_h - pad_h // 2]


def pad_same(
        x,
        kernel_size: List[int],
        stride: List[int],
        dilation: List[int] = (1, 1),
        value: float = 0,
):
    """Dynamically 'SAME'-pad a NCHW tensor (runtime cost; see module docs)."""
    ih, iw = x.size()[-2:]
    pad_h = get_same_padding(ih, kernel_size[0], stride[0], dilation[0])
    pad_w = get_same_padding(iw, kernel_size[1], stride[1], dilation[1])
    return F.pad(x, (pad_w // 2, pad_w - pad_w // 2, pad_h // 2, pad_h - pad_h // 2), value=value)


def get_padding_value(padding, kernel_size, **kwargs) -> Tuple[Union[int, List[int]], bool]:
    """Resolve a padding spec ('', 'same', 'valid', or explicit int) to
    (value, is_dynamic)."""
    dynamic = False
    if isinstance(padding, str):
        padding = padding.lower()
        if padding == 'same':
            if is_static_pad(kernel_size, **kwargs):
                padding = get_padding(kernel_size, **kwargs)
            else:
                padding = 0
                dynamic = True  # must pad at runtime per input size
        elif padding == 'valid':
            padding = 0
        else:
            # '' and anything else: PyTorch-style symmetric padding
            padding = get_padding(kernel_size, **kwargs)
    return padding, dynamic
