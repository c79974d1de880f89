"""Attention layers operating directly on NCHW feature maps.

Capability parity with reference `timm/layers/attention2d.py`:
`MultiQueryAttentionV2` (:13), `MultiQueryAttention2d` (:94, the
MobileNetV4/V5 MQA with query-stride pooling and strided depthwise K/V
downsampling) and `Attention2d` (:320, conv-projection MHSA).

On MI355X the q/k/v projections are 1x1 convs (hipBLASLt implicit GEMM);
the attention core routes through `ops.flash_attention` when shapes allow
(K/V single-head is expanded — cheap at these small map sizes).
"""
from typing import List, Optional, Type, Union

import torch
from torch import nn
from torch.nn import functional as F

from .. import ops
from .config import use_fused_attn
from .create_conv2d import create_conv2d
from .helpers import to_2tuple
from .pool2d_same import create_pool2d

__all__ = ['MultiQueryAttentionV2', 'MultiQueryAttention2d', 'Attention2d']


class MultiQueryAttentionV2(nn.Module):
    """Multi-query attention, einsum form (one shared K/V head).

    Reference `attention2d.py:13`.
    """

    def __init__(
            self,
            dim: int,
            dim_out: Optional[int] = None,
            num_heads: int = 8,
            key_dim: int = 64,
            value_dim: int = 64,
            attn_drop: float = 0.,
            proj_drop: float = 0.,
    ):
        super().__init__()
        dim_out = dim_out or dim
        self.num_heads = num_heads
        self.key_dim = key_dim
        self.value_dim = value_dim
        self.scale = key_dim ** -0.5

        self.query_proj = nn.Parameter(torch.empty(num_heads, key_dim, dim))
        self.key_proj = nn.Parameter(torch.empty(dim, key_dim))
        self.value_proj = nn.Parameter(torch.empty(dim, value_dim))
        self.attn_drop = nn.Dropout(attn_drop)
        self.out_proj = nn.Parameter(torch.empty(dim_out, num_heads, value_dim))
        self.proj_drop = nn.Dropout(proj_drop)
        self.reset_parameters()

    def reset_parameters(self):
        scale = self.key_proj.shape[0] ** -0.5
        nn.init.normal_(self.query_proj, std=scale)
        nn.init.normal_(self.key_proj, std=scale)
        nn.init.normal_(self.value_proj, std=scale)
        nn.init.normal_(self.out_proj, std=self.out_proj.shape[0] ** -0.5)

    @staticmethod
    def _flatten(t: torch.Tensor) -> torch.Tensor:
        """(B, C, ...) -> (B, N, C)."""
        return t.reshape(t.shape[0], t.shape[1], -1).transpose(1, 2)

    def forward(self, x, m: Optional[torch.Tensor] = None):
        b, _, h, w = x.shape
        m = m if m is not None else x

        xf = self._flatten(x)
        mf = self._flatten(m)

        q = torch.einsum('bnd,hkd->bnhk', xf, self.query_proj)
        k = torch.einsum('bmd,dk->bmk', mf, self.key_proj)
        attn = torch.einsum('bnhk,bmk->bnhm', q, k) * self.scale
        attn = self.attn_drop(attn.softmax(dim=-1))
        v = torch.einsum('bmd,dv->bmv', mf, self.value_proj)
        o = torch.einsum('bnhm,bmv->bnhv', attn, v)
        out = torch.einsum('bnhv,dhv->bdn', o, self.out_proj)
        out = self.proj_drop(out)
        return out.reshape(b, -1, h, w)


class MultiQueryAttention2d(nn.Module):
    """Multi-query attention with spatial down/upsampling (MobileNetV4/V5).

    Reference `attention2d.py:94`.  ``query_strides`` avg-pools queries (and
    bilinearly upsamples the output); ``kv_stride`` downsamples K/V with a
    strided depthwise conv.  K/V have ONE shared head.
    """
    fused_attn: torch.jit.Final[bool]

    def __init__(
            self,
            dim: int,
            dim_out: Optional[int] = None,
            num_heads: int = 8,
            key_dim: Optional[int] = None,
            value_dim: Optional[int] = None,
            query_strides: int = 1,
            kv_stride: int = 1,
            dw_kernel_size: int = 3,
            dilation: int = 1,
            padding: Union[str, int, List[int]] = '',
            attn_drop: float = 0.,
            proj_drop: float = 0.,
            norm_layer: Type[nn.Module] = nn.BatchNorm2d,
            use_bias: bool = False,
    ):
        super().__init__()
        dim_out = dim_out or dim
        self.num_heads = num_heads
        self.key_dim = key_dim or dim // num_heads
        self.value_dim = value_dim or dim // num_heads
        self.query_strides = to_2tuple(query_strides)
        self.kv_stride = kv_stride
        self.has_query_strides = any([s > 1 for s in self.query_strides])
        self.scale = self.key_dim ** -0.5
        self.fused_attn = use_fused_attn()
        self.drop = attn_drop

        self.query = nn.Sequential()
        if self.has_query_strides:
            if padding == 'same':
                self.query.add_module('down_pool', create_pool2d(
                    'avg', kernel_size=self.query_strides, padding='same'))
            else:
                self.query.add_module('down_pool', nn.AvgPool2d(kernel_size=query_strides))
            self.query.add_module('norm', norm_layer(dim))
        self.query.add_module('proj', create_conv2d(
            dim, self.num_heads * self.key_dim, kernel_size=1, bias=use_bias))

        self.key = nn.Sequential()
        if kv_stride > 1:
            self.key.add_module('down_conv', create_conv2d(
                dim, dim, kernel_size=dw_kernel_size, stride=kv_stride,
                dilation=dilation, padding=padding, depthwise=True))
            self.key.add_module('norm', norm_layer(dim))
        self.key.add_module('proj', create_conv2d(
            dim, self.key_dim, kernel_size=1, padding=padding, bias=use_bias))

        self.value = nn.Sequential()
        if kv_stride > 1:
            self.value.add_module('down_conv', create_conv2d(
                dim, dim, kernel_size=dw_kernel_size, stride=kv_stride,
                dilation=dilation, padding=padding, depthwise=True))
            self.value.add_module('norm', norm_layer(dim))
        self.value.add_module('proj', create_conv2d(
            dim, self.value_dim, kernel_size=1, bias=use_bias))

        self.attn_drop = nn.Dropout(attn_drop)

        self.output = nn.Sequential()
        if self.has_query_strides:
            self.output.add_module('upsample', nn.Upsample(
                scale_factor=self.query_strides, mode='bilinear', align_corners=False))
        self.output.add_module('proj', create_conv2d(
            self.value_dim * self.num_heads, dim_out, kernel_size=1, bias=use_bias))
        self.output.add_module('drop', nn.Dropout(proj_drop))

        self.init_weights()

    def init_weights(self):
        # xavier improves stability for the mobilenetv4 hybrids
        nn.init.xavier_uniform_(self.query.proj.weight)
        nn.init.xavier_uniform_(self.key.proj.weight)
        nn.init.xavier_uniform_(self.value.proj.weight)
        if self.kv_stride > 1:
            nn.init.xavier_uniform_(self.key.down_conv.weight)
            nn.init.xavier_uniform_(self.value.down_conv.weight)
        nn.init.xavier_uniform_(self.output.proj.weight)

    def forward(self, x, attn_mask: Optional[torch.Tensor] = None):
        B, C, H, W = x.shape

        q = self.query(x)       # (B, heads*k, H', W')
        q = q.reshape(B, self.num_heads, self.key_dim, -1).transpose(-1, -2)  # B, h, L, k
        k = self.key(x).reshape(B, 1, self.key_dim, -1).transpose(-1, -2)     # B, 1, P, k
        v = self.value(x).reshape(B, 1, self.value_dim, -1).transpose(-1, -2)  # B, 1, P, v

        if self.fused_attn:
            o = ops.flash_attention(
                q.contiguous(),
                k.expand(-1, self.num_heads, -1, -1).contiguous(),
                v.expand(-1, self.num_heads, -1, -1).contiguous(),
                attn_mask=attn_mask,
                dropout_p=self.attn_drop.p if self.training else 0.,
            )
        else:
            attn = (q @ k.transpose(-1, -2)) * self.scale
            if attn_mask is not None:
                attn = attn + attn_mask
            attn = self.attn_drop(attn.softmax(dim=-1))
            o = attn @ v

        # (B, h, L, v) -> (B, h*v, H', W')
        o = o.transpose(-1, -2).reshape(
            B, self.num_heads * self.value_dim,
            H // self.query_strides[0], W // self.query_strides[1])
        return self.output(o)


class Attention2d(nn.Module):
    """MHSA with 1x1-conv projections on NCHW (reference `attention2d.py:320`)."""
    fused_attn: torch.jit.Final[bool]

    def __init__(
            self,
            dim: int,
            dim_out: Optional[int] = None,
            num_heads: int = 32,
            bias: bool = True,
            expand_first: bool = False,
            head_first: bool = False,
            attn_drop: float = 0.,
            proj_drop: float = 0.,
    ):
        super().__init__()
        dim_out = dim_out or dim
        dim_attn = dim_out if expand_first else dim
        self.num_heads = num_heads
        self.dim_head = dim_attn // num_heads
        self.head_first = head_first
        self.fused_attn = use_fused_attn()

        self.qkv = nn.Conv2d(dim, dim_attn * 3, 1, bias=bias)
        self.attn_drop = nn.Dropout(attn_drop)
        self.proj = nn.Conv2d(dim_attn, dim_out, 1, bias=bias)
        self.proj_drop = nn.Dropout(proj_drop)

    def forward(self, x, attn_mask: Optional[torch.Tensor] = None):
        B, C, H, W = x.shape

        if self.head_first:
            q, k, v = self.qkv(x).view(B, self.num_heads, self.dim_head * 3, -1).chunk(3, dim=2)
        else:
            q, k, v = self.qkv(x).reshape(B, 3, self.num_heads, self.dim_head, -1).unbind(1)

        if self.fused_attn:
            x = ops.flash_attention(
                q.transpose(-1, -2).contiguous(),
                k.transpose(-1, -2).contiguous(),
                v.transpose(-1, -2).contiguous(),
                attn_mask=attn_mask,
                dropout_p=self.attn_drop.p if self.training else 0.,
            ).transpose(-1, -2).reshape(B, -1, H, W)
        else:
            q = q.transpose(-1, -2)
            v = v.transpose(-1, -2)
            attn = q @ k * q.size(-1) ** -0.5
            if attn_mask is not None:
                attn = attn + attn_mask
            attn = self.attn_drop(attn.softmax(dim=-1))
            x = (attn @ v).transpose(-1, -2).reshape(B, -1, H, W)

        x = self.proj(x)
        x = self.proj_drop(x)
        return x
