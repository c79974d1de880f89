"""CSATv2 — frequency-domain (DCT) vision model with spatial attention.

Capability parity with reference `timm/models/csatv2.py`: learnable 8x8 DCT
stem with RGB→YCbCr conversion, zigzag coefficient ordering and per-channel
frequency normalization (:199), ConvNeXt-style blocks gated by a tiny 7x7
spatial-attention transformer (:318/:411), stage-tail transformer blocks with
conv positional encoding (:435), NormMlpClassifierHead and the csatv2 /
csatv2_21m variants.

The DCT itself is two small dense GEMMs per 8x8 tile (hipBLASLt on MI355X).
"""
import math
import warnings
from functools import partial, reduce
from typing import List, Optional, Tuple, Union

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from ..layers import (
    Attention, DropPath, LayerNorm2d, LayerScale, LayerScale2d, Mlp, NormMlpClassifierHead, trunc_normal_,
)
from ..layers.grn import GlobalResponseNorm
from ._builder import build_model_with_cfg
from ._features import feature_take_indices
from ._manipulate import checkpoint, checkpoint_seq
from ._registry import generate_default_cfgs, register_model

__all__ = ['CSATv2', 'csatv2']

# DCT frequency normalization statistics (Y, Cb, Cr channels x 64 coefficients)
_DCT_MEAN = (
    (932.42657, -0.00260, 0.33415, -0.02840, 0.00003, -0.02792, -0.00183, 0.00006,
     0.00032, 0.03402, -0.00571, 0.00020, 0.00006, -0.00038, -0.00558, -0.00116,
     -0.00000, -0.00047, -0.00008, -0.00030, 0.00942, 0.00161, -0.00009, -0.00006,
     -0.00014, -0.00035, 0.00001, -0.00220, 0.00033, -0.00002, -0.00003, -0.00020,
     0.00007, -0.00000, 0.00005, 0.00293, -0.00004, 0.00006, 0.00019, 0.00004,
     0.00006, -0.00015, -0.00002, 0.00007, 0.00010, -0.00004, 0.00008, 0.00000,
     0.00008, -0.00001, 0.00015, 0.00002, 0.00007, 0.00003, 0.00004, -0.00001,
     0.00004, -0.00000, 0.00002, -0.00000, -0.00008, -0.00000, -0.00003, 0.00003),
    (962.34735, -0.00428, 0.09835, 0.00152, -0.00009, 0.00312, -0.00141, -0.00001,
     -0.00013, 0.01050, 0.00065, 0.00006, -0.00000, 0.00003, 0.00264, 0.00000,
     0.00001, 0.00007, -0.00006, 0.00003, 0.00341, 0.00163, 0.00004, 0.00003,
     -0.00001, 0.00008, -0.00000, 0.00090, 0.00018, -0.00006, -0.00001, 0.00007,
     -0.00003, -0.00001, 0.00006, 0.00084, -0.00000, -0.00001, 0.00000, 0.00004,
     -0.00001, -0.00002, 0.00000, 0.00001, 0.00002, 0.00001, 0.00004, 0.00011,
     0.00000, -0.00003, 0.00011, -0.00002, 0.00001, 0.00001, 0.00001, 0.00001,
     -0.00007, -0.00003, 0.00001, 0.00000, 0.00001, 0.00002, 0.00001, 0.00000),
    (1053.16101, -0.00213, -0.09207, 0.00186, 0.00013, 0.00034, -0.00119, 0.00002,
     0.00011, -0.00984, 0.00046, -0.00007, -0.00001, -0.00005, 0.00180, 0.00042,
     0.00002, -0.00010, 0.00004, 0.00003, -0.00301, 0.00125, -0.00002, -0.00003,
     -0.00001, -0.00001, -0.00001, 0.00056, 0.00021, 0.00001, -0.00001, 0.00002,
     -0.00001, -0.00001, 0.00005, -0.00070, -0.00002, -0.00002, 0.00005, -0.00004,
     -0.00000, 0.00002, -0.00002, 0.00001, 0.00000, -0.00003, 0.00004, 0.00007,
     0.00001, 0.00000, 0.00013, -0.00000, 0.00000, 0.00002, -0.00000, -0.00001,
     -0.00004, -0.00003, 0.00000, 0.00001, -0.00001, 0.00001, -0.00000, 0.00000),
)

_DCT_VAR = (
    (270372.37500, 6287.10645, 5974.94043, 1653.10889, 1463.91748, 1832.58997, 755.92468, 692.41528,
     648.57184, 641.46881, 285.79288, 301.62100, 380.43405, 349.84027, 374.15891, 190.30960,
     190.76746, 221.64578, 200.82646, 145.87979, 126.92046, 62.14622, 67.75562, 102.42001,
     129.74922, 130.04631, 103.12189, 97.76417, 53.17402, 54.81048, 73.48712, 81.04342,
     69.35100, 49.06024, 33.96053, 37.03279, 20.48858, 24.94830, 33.90822, 44.54912,
     47.56363, 40.03160, 30.43313, 22.63899, 26.53739, 26.57114, 21.84404, 17.41557,
     15.18253, 10.69678, 11.24111, 12.97229, 15.08971, 15.31646, 8.90409, 7.44213,
     6.66096, 6.97719, 4.17834, 3.83882, 4.51073, 2.36646, 2.41363, 1.48266),
    (18839.21094, 321.70932, 300.15259, 77.47830, 76.02293, 89.04748, 33.99642, 34.74807,
     32.12333, 28.19588, 12.04675, 14.26871, 18.45779, 16.59588, 15.67892, 7.37718,
     8.56312, 10.28946, 9.41013, 6.69090, 5.16453, 2.55186, 3.03073, 4.66765,
     5.85418, 5.74644, 4.33702, 3.66948, 1.95107, 2.26034, 3.06380, 3.50705,
     3.06359, 2.19284, 1.54454, 1.57860, 0.97078, 1.13941, 1.48653, 1.89996,
     1.95544, 1.64950, 1.24754, 0.93677, 1.09267, 1.09516, 0.94163, 0.78966,
     0.72489, 0.50841, 0.50909, 0.55664, 0.63111, 0.64125, 0.38847, 0.33378,
     0.30918, 0.33463, 0.20875, 0.19298, 0.21903, 0.13380, 0.13444, 0.09554),
    (17127.39844, 292.81421, 271.45209, 66.64056, 63.60253, 76.35437, 28.06587, 27.84831,
     25.96656, 23.60370, 9.99173, 11.34992, 14.46955, 12.92553, 12.69353, 5.91537,
     6.60187, 7.90891, 7.32825, 5.32785, 4.29660, 2.13459, 2.44135, 3.66021,
     4.50335, 4.38959, 3.34888, 2.97181, 1.60633, 1.77010, 2.35118, 2.69018,
     2.38189, 1.74596, 1.26014, 1.31684, 0.79327, 0.92046, 1.17670, 1.47609,
     1.50914, 1.28725, 0.99898, 0.74832, 0.85736, 0.85800, 0.74663, 0.63508,
     0.58748, 0.41098, 0.41121, 0.44663, 0.50277, 0.51519, 0.31729, 0.27336,
     0.25399, 0.27241, 0.17353, 0.16255, 0.18440, 0.11602, 0.11511, 0.08450),
)


def _zigzag_permutation(rows: int, cols: int) -> List[int]:
    """JPEG zigzag scan order over a rows x cols coefficient grid."""
    idx = np.arange(0, rows * cols, 1).reshape(rows, cols).tolist()
    dia = [[] for _ in range(rows + cols - 1)]
    for i in range(rows):
        for j in range(cols):
            s = i + j
            if s % 2 == 0:
                dia[s].insert(0, idx[i][j])
            else:
                dia[s].append(idx[i][j])
    zigzag = []
    for d in dia:
        zigzag.extend(d)
    return zigzag


def _dct_kernel_type_2(kernel_size: int, orthonormal: bool, device=None, dtype=None) -> torch.Tensor:
    """Type-II DCT matrix via FFT of the mirrored identity."""
    dd = dict(device=device, dtype=dtype)
    x = torch.eye(kernel_size, **dd)
    v = x.clone().contiguous().view(-1, kernel_size)
    v = torch.cat([v, v.flip([1])], dim=-1)
    v = torch.fft.fft(v, dim=-1)[:, :kernel_size]
    k = (
        torch.tensor(-1j, device=device, dtype=torch.complex64) * torch.pi
        * torch.arange(kernel_size, device=device, dtype=torch.long)[None, :]
    )
    v = (v * torch.exp(k / (kernel_size * 2))).real
    if orthonormal:
        v[:, 0] = v[:, 0] * torch.sqrt(torch.tensor(1 / (kernel_size * 4), **dd))
        v[:, 1:] = v[:, 1:] * torch.sqrt(torch.tensor(1 / (kernel_size * 2), **dd))
    return v.contiguous().view(*x.shape)


def _dct_kernel_type_3(kernel_size: int, orthonormal: bool, device=None, dtype=None) -> torch.Tensor:
    """Type-III DCT matrix (inverse of Type-II)."""
    return torch.linalg.inv(_dct_kernel_type_2(kernel_size, orthonormal, device, dtype))


class Dct1d(nn.Module):
    """1D DCT as a fixed linear transform."""

    def __init__(self, kernel_size: int, kernel_type: int = 2, orthonormal: bool = True):
        super().__init__()
        kernel = {'2': _dct_kernel_type_2, '3': _dct_kernel_type_3}
        self.register_buffer('weights', kernel[f'{kernel_type}'](kernel_size, orthonormal).T.contiguous())
        self.register_parameter('bias', None)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.linear(x, self.weights, self.bias)


class Dct2d(nn.Module):
    """Separable 2D DCT: 1D transform along both trailing dims."""

    def __init__(self, kernel_size: int, kernel_type: int = 2, orthonormal: bool = True):
        super().__init__()
        self.transform = Dct1d(kernel_size, kernel_type, orthonormal)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.transform(self.transform(x).transpose(-1, -2)).transpose(-1, -2)


def _split_out_chs(out_chs: int, ratio=(24, 4, 4)):
    g = reduce(math.gcd, ratio)
    r = tuple(x // g for x in ratio)
    denom = sum(r)
    assert out_chs % denom == 0 and out_chs >= denom, (
        f"out_chs={out_chs} can't be split into Y/Cb/Cr with ratio {ratio} "
        f'(reduced {r}); out_chs must be a multiple of {denom}.')
    unit = out_chs // denom
    y, cb, cr = (ri * unit for ri in r)
    assert y + cb + cr == out_chs and min(y, cb, cr) > 0
    return y, cb, cr


class LearnableDct2d(nn.Module):
    """8x8 DCT stem: denormalize, RGB→YCbCr, blockwise DCT, zigzag, norm,
    then per-plane 1x1 convs at a 24:4:4 channel ratio."""

    def __init__(
            self,
            kernel_size: int,
            kernel_type: int = 2,
            orthonormal: bool = True,
            out_chs: int = 32,
    ):
        super().__init__()
        self.k = kernel_size
        self.transform = Dct2d(kernel_size, kernel_type, orthonormal)
        self.permutation = _zigzag_permutation(kernel_size, kernel_size)

        y_ch, cb_ch, cr_ch = _split_out_chs(out_chs, ratio=(24, 4, 4))
        self.conv_y = nn.Conv2d(kernel_size ** 2, y_ch, kernel_size=1, padding=0)
        self.conv_cb = nn.Conv2d(kernel_size ** 2, cb_ch, kernel_size=1, padding=0)
        self.conv_cr = nn.Conv2d(kernel_size ** 2, cr_ch, kernel_size=1, padding=0)

        self.register_buffer('mean', torch.empty(3, 64), persistent=False)
        self.register_buffer('var', torch.empty(3, 64), persistent=False)
        self.register_buffer('imagenet_mean', torch.empty(3, 1, 1), persistent=False)
        self.register_buffer('imagenet_std', torch.empty(3, 1, 1), persistent=False)
        self.reset_parameters()

    def reset_parameters(self) -> None:
        self._init_buffers()

    def _init_buffers(self) -> None:
        self.mean.copy_(torch.tensor(_DCT_MEAN))
        self.var.copy_(torch.tensor(_DCT_VAR))
        self.imagenet_mean.copy_(torch.tensor([0.485, 0.456, 0.406]).view(3, 1, 1))
        self.imagenet_std.copy_(torch.tensor([0.229, 0.224, 0.225]).view(3, 1, 1))

    def _denormalize(self, x: torch.Tensor) -> torch.Tensor:
        return x.mul(self.imagenet_std).add_(self.imagenet_mean) * 255

    def _rgb_to_ycbcr(self, x: torch.Tensor) -> torch.Tensor:
        r, g, b = x[:, 0], x[:, 1], x[:, 2]
        y = r * 0.299 + g * 0.587 + b * 0.114
        cb = 0.564 * (b - y) + 128
        cr = 0.713 * (r - y) + 128
        return torch.stack([y, cb, cr], dim=1)

    def _frequency_normalize(self, x: torch.Tensor) -> torch.Tensor:
        std = self.var ** 0.5 + 1e-8
        return (x - self.mean) / std

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        b, c, h, w = x.shape
        x = self._denormalize(x)
        x = self._rgb_to_ycbcr(x)
        x = x.reshape(b, c, h // self.k, self.k, w // self.k, self.k)
        x = x.permute(0, 2, 4, 1, 3, 5)  # (B, H//k, W//k, C, k, k)
        x = self.transform(x)
        x = x.reshape(-1, c, self.k * self.k)
        x = x[:, :, self.permutation]
        x = self._frequency_normalize(x)
        x = x.reshape(b, h // self.k, w // self.k, c, -1)
        x = x.permute(0, 3, 4, 1, 2).contiguous()
        x_y = self.conv_y(x[:, 0])
        x_cb = self.conv_cb(x[:, 1])
        x_cr = self.conv_cr(x[:, 2])
        return torch.cat([x_y, x_cb, x_cr], dim=1)


class PosConv(nn.Module):
    """Depthwise-conv positional encoding over the token grid."""

    def __init__(self, in_chans: int):
        super().__init__()
        self.proj = nn.Conv2d(in_chans, in_chans, kernel_size=3, stride=1, padding=1, bias=True, groups=in_chans)

    def forward(self, x: torch.Tensor, size: Tuple[int, int]) -> torch.Tensor:
        B, N, C = x.shape
        H, W = size
        cnn_feat = x.transpose(1, 2).view(B, C, H, W)
        x = self.proj(cnn_feat) + cnn_feat
        return x.flatten(2).transpose(1, 2)


class SpatialTransformerBlock(nn.Module):
    """Tiny 1-channel single-head transformer over the 7x7 attention grid."""

    def __init__(self):
        super().__init__()
        self.pos_embed = PosConv(in_chans=1)
        self.norm1 = nn.LayerNorm(1)
        self.qkv = nn.Linear(1, 3, bias=False)
        self.norm2 = nn.LayerNorm(1)
        self.mlp = Mlp(1, 4, 1, act_layer=nn.GELU)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, C, H, W = x.shape

        shortcut = x
        x_t = x.flatten(2).transpose(1, 2)  # (B, N, 1)
        x_t = self.norm1(x_t)
        x_t = self.pos_embed(x_t, (H, W))

        qkv = self.qkv(x_t)  # (B, N, 3)
        q, k, v = qkv.unbind(-1)
        attn = (q @ k.transpose(-1, -2)).softmax(dim=-1)
        x_t = (attn @ v).unsqueeze(-1)

        x_t = x_t.transpose(1, 2).reshape(B, C, H, W)
        x = shortcut + x_t

        shortcut = x
        x_t = x.flatten(2).transpose(1, 2)
        x_t = self.mlp(self.norm2(x_t))
        x_t = x_t.transpose(1, 2).reshape(B, C, H, W)
        return shortcut + x_t


class SpatialAttention(nn.Module):
    """Spatial gate: channel avg/max stats -> 7x7 pool -> conv -> tiny transformer."""

    def __init__(self):
        super().__init__()
        self.avgpool = nn.AdaptiveAvgPool2d((7, 7))
        self.conv = nn.Conv2d(2, 1, kernel_size=7, padding=3)
        self.attn = SpatialTransformerBlock()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x_avg = x.mean(dim=1, keepdim=True)
        x_max = x.amax(dim=1, keepdim=True)
        x = torch.cat([x_avg, x_max], dim=1)
        x = self.avgpool(x)
        x = self.conv(x)
        return self.attn(x)


class Block(nn.Module):
    """ConvNeXt-style block whose output is gated by SpatialAttention."""

    def __init__(self, dim: int, drop_path: float = 0., ls_init_value: Optional[float] = None):
        super().__init__()
        self.dwconv = nn.Conv2d(dim, dim, kernel_size=7, padding=3, groups=dim)
        self.norm = nn.LayerNorm(dim, eps=1e-6)
        self.pwconv1 = nn.Linear(dim, 4 * dim)
        self.act = nn.GELU()
        self.grn = GlobalResponseNorm(4 * dim, channels_last=True)
        self.pwconv2 = nn.Linear(4 * dim, dim)
        self.ls = LayerScale2d(dim, init_values=ls_init_value) if ls_init_value else nn.Identity()
        self.drop_path = DropPath(drop_path) if drop_path > 0. else nn.Identity()
        self.attn = SpatialAttention()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shortcut = x
        x = self.dwconv(x)
        x = x.permute(0, 2, 3, 1)
        x = self.norm(x)
        x = self.pwconv1(x)
        x = self.act(x)
        x = self.grn(x)
        x = self.pwconv2(x)
        x = x.permute(0, 3, 1, 2)

        attn = self.attn(x)
        attn = F.interpolate(attn, size=x.shape[2:], mode='bilinear', align_corners=True)
        x = x * attn
        x = self.ls(x)
        return shortcut + self.drop_path(x)


class TransformerBlock(nn.Module):
    """Stage-tail transformer block with conv pos-enc and optional downsample."""

    def __init__(
            self,
            inp: int,
            oup: int,
            num_heads: int = 8,
            attn_head_dim: int = 32,
            downsample: bool = False,
            attn_drop: float = 0.,
            proj_drop: float = 0.,
            drop_path: float = 0.,
            ls_init_value: Optional[float] = None,
    ):
        super().__init__()
        hidden_dim = int(inp * 4)
        self.downsample = downsample

        if self.downsample:
            self.pool1 = nn.MaxPool2d(3, 2, 1)
            self.pool2 = nn.MaxPool2d(3, 2, 1)
            self.proj = nn.Conv2d(inp, oup, 1, 1, 0, bias=False)
        else:
            self.pool1 = nn.Identity()
            self.pool2 = nn.Identity()
            self.proj = nn.Identity()

        self.pos_embed = PosConv(in_chans=inp)
        self.norm1 = nn.LayerNorm(inp)
        self.attn = Attention(
            dim=inp,
            num_heads=num_heads,
            attn_head_dim=attn_head_dim,
            dim_out=oup,
            attn_drop=attn_drop,
            proj_drop=proj_drop,
        )
        self.ls1 = LayerScale(oup, init_values=ls_init_value) if ls_init_value else nn.Identity()
        self.drop_path1 = DropPath(drop_path) if drop_path > 0. else nn.Identity()

        self.norm2 = nn.LayerNorm(oup)
        self.mlp = Mlp(oup, hidden_dim, oup, act_layer=nn.GELU, drop=proj_drop)
        self.ls2 = LayerScale(oup, init_values=ls_init_value) if ls_init_value else nn.Identity()
        self.drop_path2 = DropPath(drop_path) if drop_path > 0. else nn.Identity()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.downsample:
            shortcut = self.proj(self.pool1(x))
            x_t = self.pool2(x)
        else:
            shortcut = x
            x_t = x
        B, C, H, W = x_t.shape
        x_t = x_t.flatten(2).transpose(1, 2)
        x_t = self.norm1(x_t)
        x_t = self.pos_embed(x_t, (H, W))
        x_t = self.ls1(self.attn(x_t))
        x_t = x_t.transpose(1, 2).reshape(B, -1, H, W)
        x = shortcut + self.drop_path1(x_t)

        B, C, H, W = x.shape
        shortcut = x
        x_t = x.flatten(2).transpose(1, 2)
        x_t = self.ls2(self.mlp(self.norm2(x_t)))
        x_t = x_t.transpose(1, 2).reshape(B, C, H, W)
        return shortcut + self.drop_path2(x_t)


class CSATv2(nn.Module):
    """CSATv2 (reference `csatv2.py:538`)."""

    def __init__(
            self,
            num_classes: int = 1000,
            in_chans: int = 3,
            dims: Tuple[int, ...] = (32, 72, 168, 386),
            depths: Tuple[int, ...] = (2, 2, 8, 6),
            transformer_depths: Tuple[int, ...] = (0, 0, 2, 2),
            drop_path_rate: float = 0.0,
            transformer_drop_path: bool = False,
            ls_init_value: Optional[float] = None,
            global_pool: str = 'avg',
            **kwargs,
    ) -> None:
        super().__init__()
        if in_chans != 3:
            warnings.warn(
                f'CSATv2 is designed for 3-channel RGB input. '
                f'in_chans={in_chans} may not work correctly with the DCT stem.')
        self.num_classes = num_classes
        self.in_chans = in_chans
        self.global_pool = global_pool
        self.grad_checkpointing = False

        self.num_features = dims[-1]
        self.head_hidden_size = self.num_features

        self.feature_info = [dict(num_chs=dims[0], reduction=8, module='stem_dct')]
        reduction = 8
        for i, dim in enumerate(dims):
            if i > 0:
                reduction *= 2
            self.feature_info.append(dict(num_chs=dim, reduction=reduction, module=f'stages.{i}'))

        # drop path for conv blocks only unless transformer_drop_path
        total_blocks = sum(depths) if transformer_drop_path else sum(d - t for d, t in zip(depths, transformer_depths))
        dp_iter = iter(torch.linspace(0, drop_path_rate, total_blocks).tolist())
        dp_rates = []
        for depth, t_depth in zip(depths, transformer_depths):
            dp_rates += [next(dp_iter) for _ in range(depth - t_depth)]
            dp_rates += [next(dp_iter) if transformer_drop_path else 0. for _ in range(t_depth)]

        self.stem_dct = LearnableDct2d(8, out_chs=dims[0])

        dp_iter = iter(dp_rates)
        stages = []
        for i, (dim, depth, t_depth) in enumerate(zip(dims, depths, transformer_depths)):
            layers = (
                ([nn.Conv2d(dims[i - 1], dim, kernel_size=2, stride=2)] if i > 0 else []) +
                [Block(dim=dim, drop_path=next(dp_iter), ls_init_value=ls_init_value)
                 for _ in range(depth - t_depth)] +
                [TransformerBlock(inp=dim, oup=dim, drop_path=next(dp_iter), ls_init_value=ls_init_value)
                 for _ in range(t_depth)] +
                ([LayerNorm2d(dim, eps=1e-6)] if i < len(depths) - 1 else [])
            )
            stages.append(nn.Sequential(*layers))
        self.stages = nn.Sequential(*stages)

        self.head = NormMlpClassifierHead(dims[-1], num_classes, pool_type=global_pool)

        self.init_weights(needs_reset=False)

    def init_weights(self, needs_reset: bool = True):
        self.apply(partial(self._init_weights, needs_reset=needs_reset))

    def _init_weights(self, m: nn.Module, needs_reset: bool = True) -> None:
        if isinstance(m, (nn.Conv2d, nn.Linear)):
            trunc_normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)
        elif needs_reset and hasattr(m, 'reset_parameters'):
            m.reset_parameters()

    @torch.jit.ignore
    def get_classifier(self) -> nn.Module:
        return self.head.fc

    def reset_classifier(self, num_classes: int, global_pool: Optional[str] = None) -> None:
        self.num_classes = num_classes
        if global_pool is not None:
            self.global_pool = global_pool
        self.head.reset(num_classes, pool_type=global_pool)

    @torch.jit.ignore
    def set_grad_checkpointing(self, enable: bool = True) -> None:
        self.grad_checkpointing = enable

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        x = self.stem_dct(x)
        if self.grad_checkpointing and not torch.jit.is_scripting():
            x = checkpoint_seq(self.stages, x)
        else:
            x = self.stages(x)
        return x

    def forward_intermediates(
            self,
            x: torch.Tensor,
            indices: Optional[Union[int, List[int]]] = None,
            norm: bool = False,
            stop_early: bool = False,
            output_fmt: str = 'NCHW',
            intermediates_only: bool = False,
    ) -> Union[List[torch.Tensor], Tuple[torch.Tensor, List[torch.Tensor]]]:
        assert output_fmt == 'NCHW', 'Output format must be NCHW.'
        intermediates = []
        # 5 feature levels: stem_dct (0) + stages 0-3 (1-4)
        take_indices, max_index = feature_take_indices(len(self.stages) + 1, indices)

        x = self.stem_dct(x)
        if 0 in take_indices:
            intermediates.append(x)

        if torch.jit.is_scripting() or not stop_early:
            stages = self.stages
        else:
            stages = self.stages[:max_index] if max_index > 0 else []

        for feat_idx, stage in enumerate(stages):
            if self.grad_checkpointing and not torch.jit.is_scripting():
                x = checkpoint(stage, x)
            else:
                x = stage(x)
            if feat_idx + 1 in take_indices:
                intermediates.append(x)

        if intermediates_only:
            return intermediates
        return x, intermediates

    def prune_intermediate_layers(
            self,
            indices: Union[int, List[int]] = 1,
            prune_norm: bool = False,
            prune_head: bool = True,
    ) -> List[int]:
        take_indices, max_index = feature_take_indices(len(self.stages) + 1, indices)
        self.stages = self.stages[:max_index] if max_index > 0 else nn.Sequential()
        if prune_norm:
            self.head.norm = nn.Identity()
        if prune_head:
            self.reset_classifier(0, '')
        return take_indices

    def forward_head(self, x: torch.Tensor, pre_logits: bool = False) -> torch.Tensor:
        return self.head(x, pre_logits=pre_logits)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.forward_features(x)
        return self.forward_head(x)


def _cfg(url='', **kwargs):
    return {
        'url': url,
        'num_classes': 1000, 'input_size': (3, 512, 512), 'pool_size': (8, 8),
        'mean': (0.485, 0.456, 0.406), 'std': (0.229, 0.224, 0.225),
        'interpolation': 'bilinear', 'crop_pct': 1.0,
        'classifier': 'head.fc', 'first_conv': [],
        **kwargs,
    }


default_cfgs = generate_default_cfgs({
    'csatv2.r512_in1k': _cfg(hf_hub_id='timm/'),
    'csatv2_21m.sw_r640_in1k': _cfg(hf_hub_id='timm/', input_size=(3, 640, 640), interpolation='bicubic'),
    'csatv2_21m.sw_r512_in1k': _cfg(hf_hub_id='timm/', pool_size=(10, 10), interpolation='bicubic'),
})


def _create_csatv2(variant: str, pretrained: bool = False, **kwargs) -> CSATv2:
    out_indices = kwargs.pop('out_indices', (1, 2, 3, 4))
    return build_model_with_cfg(
        CSATv2,
        variant,
        pretrained,
        feature_cfg=dict(out_indices=out_indices, flatten_sequential=True),
        **kwargs,
    )


@register_model
def csatv2(pretrained: bool = False, **kwargs) -> CSATv2:
    return _create_csatv2('csatv2', pretrained, **kwargs)


@register_model
def csatv2_21m(pretrained: bool = False, **kwargs) -> CSATv2:
    model_args = dict(
        dims=(48, 96, 224, 448),
        depths=(3, 3, 10, 8),
        transformer_depths=(0, 0, 4, 3),
    )
    return _create_csatv2('csatv2_21m', pretrained, **dict(model_args, **kwargs))
