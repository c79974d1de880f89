"""Convolutional self-attention layers for hybrid CNN/attention backbones.

Capability parity with reference `timm/layers/bottleneck_attn.py` (BoTNet),
`timm/layers/halo_attn.py` (HaloNet) and `timm/layers/lambda_layer.py`
(LambdaNetworks).  All three share the decomposed 2D relative-position logit
scheme from "Attention Augmented Convolutional Networks"; here that is one
module (``RelPos2d``) parameterised by query-block and key-window size rather
than two near-duplicate per-file implementations.

These layers slot into ByobNet ``self_attn`` blocks (see models/byoanet.py).
On MI355X the batched-matmul attention paths lower onto hipBLASLt strided
GEMMs; the window sizes are small (8..23) so these stay bandwidth-friendly.
"""
from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from .grid import ndgrid
from .helpers import to_2tuple, make_divisible
from .trace_utils import _assert
from .weight_init import trunc_normal_

__all__ = ['RelPos2d', 'BottleneckAttn', 'HaloAttn', 'LambdaLayer']


def _skew_rel_logits(q: torch.Tensor, rel_k: torch.Tensor) -> torch.Tensor:
    """Relative→absolute logits along one axis via the pad/reshape skew trick.

    Args:
        q: (N, W, d) queries for one axis (W query positions).
        rel_k: (R, d) learned relative keys, R = 2 * K - 1 for K key positions.
    Returns:
        (N, W, K) logits, entry [n, i, j] = q[n, i] . rel_k[j - i + K - 1].
    """
    N, W, _ = q.shape
    R = rel_k.shape[0]
    K = (R + 1) // 2
    t = q @ rel_k.transpose(0, 1)                 # (N, W, R) relative-indexed
    t = F.pad(t, (0, 1)).reshape(N, W * (R + 1))  # one pad col per row shifts
    t = F.pad(t, (0, R - W)).reshape(N, W + 1, R)  # rows now absolute-aligned
    return t[:, :W, K - 1:]


class RelPos2d(nn.Module):
    """Decomposed (height + width) 2D relative position logits.

    Generalises reference `bottleneck_attn.py:56` (q block == k window ==
    feature map) and `halo_attn.py:61` (q block smaller than k window).
    Parameter names/shapes (`height_rel`, `width_rel` of (2*k-1, d)) match the
    reference so pretrained checkpoints map 1:1.
    """

    def __init__(
            self,
            q_size,
            k_size=None,
            dim_head: int = 64,
            scale: float = 1.0,
    ):
        super().__init__()
        self.q_h, self.q_w = to_2tuple(q_size)
        self.k_h, self.k_w = to_2tuple(k_size if k_size is not None else q_size)
        self.dim_head = dim_head
        self.scale = scale
        self.height_rel = nn.Parameter(torch.empty(2 * self.k_h - 1, dim_head))
        self.width_rel = nn.Parameter(torch.empty(2 * self.k_w - 1, dim_head))
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.normal_(self.height_rel, std=self.scale)
        nn.init.normal_(self.width_rel, std=self.scale)

    def forward(self, q: torch.Tensor) -> torch.Tensor:
        """q: (..., q_h * q_w, d) → logits (..., q_h * q_w, k_h * k_w)."""
        lead = q.shape[:-2]
        d = q.shape[-1]
        q = q.reshape(-1, self.q_h, self.q_w, d)
        B = q.shape[0]

        # width axis: same logits for every key row -> expand over k_h
        lw = _skew_rel_logits(q.reshape(-1, self.q_w, d), self.width_rel)
        lw = lw.reshape(B, self.q_h, 1, self.q_w, self.k_w).expand(-1, -1, self.k_h, -1, -1)
        lw = lw.permute(0, 1, 3, 2, 4)  # (B, q_h, q_w, k_h, k_w)

        # height axis: transpose H<->W, skew, expand over k_w
        qt = q.transpose(1, 2).reshape(-1, self.q_h, d)
        lh = _skew_rel_logits(qt, self.height_rel)
        lh = lh.reshape(B, self.q_w, 1, self.q_h, self.k_h).expand(-1, -1, self.k_w, -1, -1)
        lh = lh.permute(0, 3, 1, 4, 2)  # (B, q_h, q_w, k_h, k_w)

        out = (lh + lw).reshape(*lead, self.q_h * self.q_w, self.k_h * self.k_w)
        return out


class BottleneckAttn(nn.Module):
    """Global 2D self-attention w/ relative position (BoTNet).

    Reference `timm/layers/bottleneck_attn.py:100`.  Requires a concrete
    ``feat_size`` (fixed input size models).  Output dim set by ``dim_out``;
    q/k head dim from ``dim_head`` or ``dim_out * qk_ratio // num_heads``.
    """

    def __init__(
            self,
            dim: int,
            dim_out: Optional[int] = None,
            feat_size: Optional[Tuple[int, int]] = None,
            stride: int = 1,
            num_heads: int = 4,
            dim_head: Optional[int] = None,
            qk_ratio: float = 1.0,
            qkv_bias: bool = False,
            scale_pos_embed: bool = False,
    ):
        super().__init__()
        assert feat_size is not None, 'BottleneckAttn requires a fixed feat_size'
        dim_out = dim_out or dim
        assert dim_out % num_heads == 0
        self.num_heads = num_heads
        self.dim_head_qk = dim_head or make_divisible(dim_out * qk_ratio, divisor=8) // num_heads
        self.dim_head_v = dim_out // num_heads
        self.dim_out_qk = num_heads * self.dim_head_qk
        self.dim_out_v = num_heads * self.dim_head_v
        self.scale = self.dim_head_qk ** -0.5
        self.scale_pos_embed = scale_pos_embed

        self.qkv = nn.Conv2d(dim, self.dim_out_qk * 2 + self.dim_out_v, 1, bias=qkv_bias)
        self.pos_embed = RelPos2d(feat_size, dim_head=self.dim_head_qk, scale=self.scale)
        self.pool = nn.AvgPool2d(2, 2) if stride == 2 else nn.Identity()
        self.reset_parameters()

    def reset_parameters(self):
        trunc_normal_(self.qkv.weight, std=self.qkv.weight.shape[1] ** -0.5)
        trunc_normal_(self.pos_embed.height_rel, std=self.scale)
        trunc_normal_(self.pos_embed.width_rel, std=self.scale)

    def forward(self, x):
        B, C, H, W = x.shape
        _assert(H == self.pos_embed.q_h, '')
        _assert(W == self.pos_embed.q_w, '')

        qkv = self.qkv(x)
        q, k, v = qkv.split([self.dim_out_qk, self.dim_out_qk, self.dim_out_v], dim=1)
        # heads folded into batch; k left (d, N) for the q @ k logits GEMM
        q = q.reshape(B * self.num_heads, self.dim_head_qk, -1).transpose(-1, -2)
        k = k.reshape(B * self.num_heads, self.dim_head_qk, -1)
        v = v.reshape(B * self.num_heads, self.dim_head_v, -1).transpose(-1, -2)

        if self.scale_pos_embed:
            attn = (q @ k + self.pos_embed(q)) * self.scale
        else:
            attn = (q @ k) * self.scale + self.pos_embed(q)
        attn = attn.softmax(dim=-1)

        out = (attn @ v).transpose(-1, -2).reshape(B, self.dim_out_v, H, W)
        return self.pool(out)


class HaloAttn(nn.Module):
    """Blocked local self-attention w/ halo overlap (HaloNet).

    Reference `timm/layers/halo_attn.py:114`.  Queries come from
    non-overlapping ``block_size`` blocks; keys/values from windows of
    ``block_size + 2 * halo_size`` gathered with strided ``unfold``.
    """

    def __init__(
            self,
            dim: int,
            dim_out: Optional[int] = None,
            feat_size: Optional[Tuple[int, int]] = None,  # unused, block cfg compat
            stride: int = 1,
            num_heads: int = 8,
            dim_head: Optional[int] = None,
            block_size: int = 8,
            halo_size: int = 3,
            qk_ratio: float = 1.0,
            qkv_bias: bool = False,
            avg_down: bool = False,
            scale_pos_embed: bool = False,
    ):
        super().__init__()
        dim_out = dim_out or dim
        assert dim_out % num_heads == 0
        assert stride in (1, 2)
        self.num_heads = num_heads
        self.dim_head_qk = dim_head or make_divisible(dim_out * qk_ratio, divisor=8) // num_heads
        self.dim_head_v = dim_out // num_heads
        self.dim_out_qk = num_heads * self.dim_head_qk
        self.dim_out_v = num_heads * self.dim_head_v
        self.scale = self.dim_head_qk ** -0.5
        self.scale_pos_embed = scale_pos_embed
        self.block_size = self.block_size_ds = block_size
        self.halo_size = halo_size
        self.win_size = block_size + halo_size * 2
        self.block_stride = 1
        use_avg_pool = False
        if stride > 1:
            use_avg_pool = avg_down or block_size % stride != 0
            self.block_stride = 1 if use_avg_pool else stride
            self.block_size_ds = self.block_size // self.block_stride

        self.q = nn.Conv2d(dim, self.dim_out_qk, 1, stride=self.block_stride, bias=qkv_bias)
        self.kv = nn.Conv2d(dim, self.dim_out_qk + self.dim_out_v, 1, bias=qkv_bias)
        self.pos_embed = RelPos2d(
            self.block_size_ds, self.win_size, dim_head=self.dim_head_qk, scale=self.scale)
        self.pool = nn.AvgPool2d(2, 2) if use_avg_pool else nn.Identity()
        self.reset_parameters()

    def reset_parameters(self):
        std = self.q.weight.shape[1] ** -0.5
        trunc_normal_(self.q.weight, std=std)
        trunc_normal_(self.kv.weight, std=std)
        trunc_normal_(self.pos_embed.height_rel, std=self.scale)
        trunc_normal_(self.pos_embed.width_rel, std=self.scale)

    def forward(self, x):
        B, C, H, W = x.shape
        _assert(H % self.block_size == 0, '')
        _assert(W % self.block_size == 0, '')
        nh_blocks = H // self.block_size
        nw_blocks = W // self.block_size
        num_blocks = nh_blocks * nw_blocks
        bs_ds = self.block_size_ds

        # queries per block: (B*heads, num_blocks, bs_ds^2, d_qk)
        q = self.q(x)
        q = q.reshape(-1, self.dim_head_qk, nh_blocks, bs_ds, nw_blocks, bs_ds)
        q = q.permute(0, 1, 3, 5, 2, 4).reshape(
            B * self.num_heads, self.dim_head_qk, -1, num_blocks).transpose(1, 3)

        # keys/values from halo windows: pad then double-unfold into
        # (B*heads, num_blocks, win^2, d) overlapping windows
        kv = self.kv(x)
        kv = F.pad(kv, [self.halo_size] * 4)
        kv = kv.unfold(2, self.win_size, self.block_size).unfold(3, self.win_size, self.block_size)
        kv = kv.reshape(
            B * self.num_heads, self.dim_head_qk + self.dim_head_v, num_blocks, -1).permute(0, 2, 3, 1)
        k, v = kv.split([self.dim_head_qk, self.dim_head_v], dim=-1)

        if self.scale_pos_embed:
            attn = (q @ k.transpose(-1, -2) + self.pos_embed(q)) * self.scale
        else:
            attn = (q @ k.transpose(-1, -2)) * self.scale + self.pos_embed(q)
        attn = attn.softmax(dim=-1)

        out = (attn @ v).transpose(1, 3)  # (B*heads, d_v, bs_ds^2, num_blocks)
        out = out.reshape(-1, bs_ds, bs_ds, nh_blocks, nw_blocks)
        out = out.permute(0, 3, 1, 4, 2).contiguous().view(
            B, self.dim_out_v, H // self.block_stride, W // self.block_stride)
        return self.pool(out)


def _lambda_rel_indices(size, device=None):
    """(2, M, M) table of relative offsets, shifted non-negative."""
    size = to_2tuple(size)
    pos = torch.stack(ndgrid(
        torch.arange(size[0], device=device, dtype=torch.long),
        torch.arange(size[1], device=device, dtype=torch.long),
    )).flatten(1)
    rel = pos[:, None, :] - pos[:, :, None]
    rel[0] += size[0] - 1
    rel[1] += size[1] - 1
    return rel


class LambdaLayer(nn.Module):
    """Lambda layer (LambdaNetworks) — content + position lambdas.

    Reference `timm/layers/lambda_layer.py:46`.  ``r`` set → local 3D-conv
    position lambdas (input-size agnostic); ``r=None`` → relative position
    embedding (needs ``feat_size``).  Intra-depth u fixed at 1.
    """

    def __init__(
            self,
            dim: int,
            dim_out: Optional[int] = None,
            feat_size: Optional[Tuple[int, int]] = None,
            stride: int = 1,
            num_heads: int = 4,
            dim_head: int = 16,
            r: Optional[int] = 9,
            qk_ratio: float = 1.0,
            qkv_bias: bool = False,
    ):
        super().__init__()
        dim_out = dim_out or dim
        assert dim_out % num_heads == 0, 'dim_out must be divisible by num_heads'
        self.dim_qk = dim_head or make_divisible(dim_out * qk_ratio, divisor=8) // num_heads
        self.num_heads = num_heads
        self.dim_v = dim_out // num_heads

        self.qkv = nn.Conv2d(
            dim, num_heads * self.dim_qk + self.dim_qk + self.dim_v, kernel_size=1, bias=qkv_bias)
        self.norm_q = nn.BatchNorm2d(num_heads * self.dim_qk)
        self.norm_v = nn.BatchNorm2d(self.dim_v)

        if r is not None:
            # position lambdas from a local (r x r) conv over v
            self.conv_lambda = nn.Conv3d(1, self.dim_qk, (r, r, 1), padding=(r // 2, r // 2, 0))
            self.pos_emb = None
            self.rel_pos_indices = None
            self.feat_size = None
        else:
            assert feat_size is not None
            feat_size = to_2tuple(feat_size)
            self.feat_size = feat_size
            rel_size = [2 * s - 1 for s in feat_size]
            M = feat_size[0] * feat_size[1]
            self.conv_lambda = None
            self.pos_emb = nn.Parameter(torch.empty(rel_size[0], rel_size[1], self.dim_qk))
            self.register_buffer(
                'rel_pos_indices', torch.empty(2, M, M, dtype=torch.long), persistent=False)

        self.pool = nn.AvgPool2d(2, 2) if stride == 2 else nn.Identity()
        self.reset_parameters()

    def reset_parameters(self):
        trunc_normal_(self.qkv.weight, std=self.qkv.weight.shape[1] ** -0.5)
        if self.conv_lambda is not None:
            trunc_normal_(self.conv_lambda.weight, std=self.dim_qk ** -0.5)
        if self.pos_emb is not None:
            trunc_normal_(self.pos_emb, std=.02)
        if self.rel_pos_indices is not None:
            self.rel_pos_indices.copy_(
                _lambda_rel_indices(self.feat_size, device=self.rel_pos_indices.device))

    def forward(self, x):
        B, C, H, W = x.shape
        M = H * W
        qkv = self.qkv(x)
        q, k, v = qkv.split([self.num_heads * self.dim_qk, self.dim_qk, self.dim_v], dim=1)
        q = self.norm_q(q).reshape(B, self.num_heads, self.dim_qk, M).transpose(-1, -2)
        v = self.norm_v(v).reshape(B, self.dim_v, M).transpose(-1, -2)   # B, M, V
        k = F.softmax(k.reshape(B, self.dim_qk, M), dim=-1)              # B, K, M

        content_lam = k @ v                                # B, K, V
        content_out = q @ content_lam.unsqueeze(1)         # B, heads, M, V

        if self.pos_emb is None:
            pos_lam = self.conv_lambda(v.reshape(B, 1, H, W, self.dim_v))
            pos_lam = pos_lam.reshape(B, 1, self.dim_qk, H * W, self.dim_v).transpose(2, 3)
        else:
            emb = self.pos_emb[self.rel_pos_indices[0], self.rel_pos_indices[1]].expand(B, -1, -1, -1)
            pos_lam = (emb.transpose(-1, -2) @ v.unsqueeze(1)).unsqueeze(1)  # B, 1, M, K, V
        position_out = (q.unsqueeze(-2) @ pos_lam).squeeze(-2)               # B, heads, M, V

        out = (content_out + position_out).transpose(-1, -2).reshape(B, self.num_heads * self.dim_v, H, W)
        return self.pool(out)
