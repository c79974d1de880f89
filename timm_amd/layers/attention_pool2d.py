"""Attention-based 2D feature pooling (CLIP-ResNet heads).

Capability parity with reference `timm/layers/attention_pool2d.py`:
`RotAttentionPool2d` (:22, rotary rel-pos — resolution-agnostic) and
`AttentionPool2d` (:174, learned absolute pos embed, resampled on size
change).  Both replace spatial average pooling with one MHSA step whose
query is the mean (or a learned class) token; used as ByobNet head types
for the CLIP ResNet variants.

The single attention step runs through `ops.flash_attention` on MI355X.
"""
from typing import Optional, Tuple, Union

import torch
import torch.nn as nn

from .. import ops
from .config import use_fused_attn
from .helpers import to_2tuple
from .pos_embed import resample_abs_pos_embed
from .pos_embed_sincos import apply_rot_embed_cat, create_rope_embed
from .weight_init import trunc_normal_

__all__ = ['RotAttentionPool2d', 'AttentionPool2d']


class _AttnPoolBase(nn.Module):
    """Shared projection / pooling plumbing for both attention pools."""
    fused_attn: torch.jit.Final[bool]

    def __init__(
            self,
            in_features: int,
            out_features: Optional[int] = None,
            embed_dim: Optional[int] = None,
            head_dim: Optional[int] = 64,
            num_heads: Optional[int] = None,
            qkv_bias: bool = True,
            qkv_separate: bool = False,
            pool_type: str = 'token',
            class_token: bool = False,
            drop_rate: float = 0.,
    ):
        super().__init__()
        assert pool_type in ('', 'token')
        self.embed_dim = embed_dim = embed_dim or in_features
        self.in_features = in_features
        if out_features is None:
            self.out_features = in_features
        elif out_features > 0:
            self.out_features = out_features
        else:
            self.out_features = embed_dim  # out_features=0 disables projection
        if num_heads is not None:
            assert embed_dim % num_heads == 0
            head_dim = embed_dim // num_heads
        else:
            assert embed_dim % head_dim == 0
            num_heads = embed_dim // head_dim
        self.num_heads = num_heads
        self.head_dim = head_dim
        self.pool_type = pool_type.lower()
        self.scale = self.head_dim ** -0.5
        self.fused_attn = use_fused_attn()

        if class_token:
            self.cls_token = nn.Parameter(torch.zeros(1, embed_dim))
        else:
            self.cls_token = None

        if qkv_separate:
            self.q = nn.Linear(in_features, embed_dim, bias=qkv_bias)
            self.k = nn.Linear(in_features, embed_dim, bias=qkv_bias)
            self.v = nn.Linear(in_features, embed_dim, bias=qkv_bias)
            self.qkv = None
        else:
            self.q = self.k = self.v = None
            self.qkv = nn.Linear(in_features, embed_dim * 3, bias=qkv_bias)
        self.drop = nn.Dropout(drop_rate)
        self.proj = nn.Linear(embed_dim, self.out_features) if out_features != 0 else nn.Identity()

    def init_weights(self, zero_init_last: bool = False):
        if self.qkv is None:
            in_features = self.q.in_features
            for lin in (self.q, self.k, self.v):
                trunc_normal_(lin.weight, std=in_features ** -0.5)
                nn.init.zeros_(lin.bias)
        else:
            trunc_normal_(self.qkv.weight, std=self.qkv.in_features ** -0.5)
            nn.init.zeros_(self.qkv.bias)

    def reset(self, num_classes: Optional[int] = None, pool_type: Optional[str] = None):
        # used as a model head, so needs a compatible reset()
        if pool_type is not None:
            assert pool_type in ('', 'token')
            self.pool_type = pool_type
        if num_classes is not None:
            self.proj = nn.Linear(self.embed_dim, num_classes) if num_classes > 0 else nn.Identity()
            self.out_features = num_classes if num_classes > 0 else self.embed_dim

    def _prepend_token(self, x: torch.Tensor) -> torch.Tensor:
        if self.cls_token is None:
            return torch.cat([x.mean(1, keepdim=True), x], dim=1)
        return torch.cat([self.cls_token.expand(x.shape[0], -1, -1), x], dim=1)

    def _qkv(self, x: torch.Tensor, B: int, L: int):
        if self.qkv is None:
            q = self.q(x).reshape(B, L, self.num_heads, self.head_dim).transpose(1, 2)
            k = self.k(x).reshape(B, L, self.num_heads, self.head_dim).transpose(1, 2)
            v = self.v(x).reshape(B, L, self.num_heads, self.head_dim).transpose(1, 2)
            return q, k, v
        qkv = self.qkv(x).reshape(B, L, 3, self.num_heads, self.head_dim).permute(2, 0, 3, 1, 4)
        return qkv.unbind(0)

    def _attend(self, q, k, v):
        if self.fused_attn:
            return ops.flash_attention(q, k, v)
        q = q * self.scale
        attn = (q @ k.transpose(-2, -1)).softmax(dim=-1)
        return attn @ v

    def _pool(self, x: torch.Tensor, H: int, W: int) -> torch.Tensor:
        if self.pool_type == 'token':
            return x[:, 0]
        # not pooled: spatial output without token
        return x[:, 1:].reshape(x.shape[0], H, W, -1).permute(0, 3, 1, 2)


class RotAttentionPool2d(_AttnPoolBase):
    """Attention pool with rotary relative position embedding (size-agnostic)."""

    def __init__(
            self,
            in_features: int,
            out_features: Optional[int] = None,
            ref_feat_size: Union[int, Tuple[int, int]] = 7,
            embed_dim: Optional[int] = None,
            head_dim: Optional[int] = 64,
            num_heads: Optional[int] = None,
            qkv_bias: bool = True,
            qkv_separate: bool = False,
            pool_type: str = 'token',
            class_token: bool = False,
            drop_rate: float = 0.,
            rope_type: str = 'cat',
    ):
        super().__init__(
            in_features, out_features=out_features, embed_dim=embed_dim,
            head_dim=head_dim, num_heads=num_heads, qkv_bias=qkv_bias,
            qkv_separate=qkv_separate, pool_type=pool_type,
            class_token=class_token, drop_rate=drop_rate)
        self.rope_type = rope_type
        self.pos_embed = create_rope_embed(
            rope_type=rope_type,
            dim=self.embed_dim,
            num_heads=self.num_heads,
            in_pixels=False,
            ref_feat_shape=to_2tuple(ref_feat_size),
            rotate_half=False,
        )

    def forward(self, x, pre_logits: bool = False):
        B, _, H, W = x.shape
        N = H * W
        x = x.flatten(2).transpose(1, 2)
        x = self._prepend_token(x)
        q, k, v = self._qkv(x, B, N + 1)

        rope = self.pos_embed.get_embed((H, W))
        if isinstance(rope, tuple):
            rope = torch.cat(rope, dim=-1)
        # the pooled token (index 0) gets no rotation
        q = torch.cat([q[:, :, :1, :], apply_rot_embed_cat(q[:, :, 1:, :], rope)], dim=2).type_as(v)
        k = torch.cat([k[:, :, :1, :], apply_rot_embed_cat(k[:, :, 1:, :], rope)], dim=2).type_as(v)

        x = self._attend(q, k, v)
        x = x.transpose(1, 2).reshape(B, N + 1, -1)
        x = self.drop(x)
        if pre_logits:
            return self._pool(x, H, W)
        x = self.proj(x)
        return self._pool(x, H, W)


class AttentionPool2d(_AttnPoolBase):
    """Attention pool with learned absolute position embedding (CLIP style)."""

    def __init__(
            self,
            in_features: int,
            feat_size: Union[int, Tuple[int, int]] = 7,
            out_features: Optional[int] = None,
            embed_dim: Optional[int] = None,
            head_dim: Optional[int] = 64,
            num_heads: Optional[int] = None,
            qkv_bias: bool = True,
            qkv_separate: bool = False,
            pool_type: str = 'token',
            class_token: bool = False,
            drop_rate: float = 0.,
    ):
        super().__init__(
            in_features, out_features=out_features, embed_dim=embed_dim,
            head_dim=head_dim, num_heads=num_heads, qkv_bias=qkv_bias,
            qkv_separate=qkv_separate, pool_type=pool_type,
            class_token=class_token, drop_rate=drop_rate)
        self.feat_size = to_2tuple(feat_size)
        self.seq_len = self.feat_size[0] * self.feat_size[1]
        self.pos_embed = nn.Parameter(torch.zeros(self.seq_len + 1, in_features))
        self.init_weights()

    def init_weights(self, zero_init_last: bool = False):
        super().init_weights(zero_init_last=zero_init_last)
        trunc_normal_(self.pos_embed, std=self.in_features ** -0.5)

    def forward(self, x, pre_logits: bool = False):
        B, _, H, W = x.shape
        N = H * W
        x = x.flatten(2).transpose(1, 2)
        x = self._prepend_token(x)
        pos_embed = resample_abs_pos_embed(self.pos_embed.unsqueeze(0), (H, W), num_prefix_tokens=1)
        x = x + pos_embed

        q, k, v = self._qkv(x, B, N + 1)
        x = self._attend(q, k, v)
        x = x.transpose(1, 2).reshape(B, N + 1, -1)
        x = self.drop(x)
        if pre_logits:
            return self._pool(x, H, W)
        x = self.proj(x)
        return self._pool(x, H, W)
