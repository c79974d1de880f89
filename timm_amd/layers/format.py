"""Tensor layout tags and conversions (reference `timm/layers/format.py`)."""
from enum import Enum
from typing import Union

import torch


class Format(str, Enum):
    NCHW = 'NCHW'
    NHWC = 'NHWC'
    NCL = 'NCL'
    NLC = 'NLC'


FormatT = Union[str, Format]

_SPATIAL_DIMS = {
    Format.NLC: (1,),
    Format.NCL: (2,),
    Format.NHWC: (1, 2),
    Format.NCHW: (2, 3),
}

_CHANNEL_DIM = {
    Format.NHWC: 3,
    Format.NLC: 2,
    Format.NCL: 1,
    Format.NCHW: 1,
}


def get_spatial_dim(fmt: FormatT):
    return _SPATIAL_DIMS[Format(fmt)]


def get_channel_dim(fmt: FormatT):
    return _CHANNEL_DIM[Format(fmt)]


def nchw_to(x: torch.Tensor, fmt: Format):
    if fmt == Format.NHWC:
        return x.permute(0, 2, 3, 1)
    if fmt == Format.NLC:
        return x.flatten(2).transpose(1, 2)
    if fmt == Format.NCL:
        return x.flatten(2)
    return x


def nhwc_to(x: torch.Tensor, fmt: Format):
    if fmt == Format.NCHW:
        return x.permute(0, 3, 1, 2)
    if fmt == Format.NLC:
        return x.flatten(1, 2)
    if fmt == Format.NCL:
        return x.flatten(1, 2).transpose(1, 2)
    return x
