"""Sin-cos / Fourier / rotary (RoPE) position embeddings.

Capability parity with reference `timm/layers/pos_embed_sincos.py` (1,357 LoC):
`build_sincos2d_pos_embed` (:39), `build_fourier_pos_embed` (:86),
`rot`/`apply_rot_embed_cat` (:228-297), `RotaryEmbedding` (:393),
`RotaryEmbeddingCat` (:534 — used by EVA02), factory `create_rope_embed` (:1315).

On device the RoPE apply (x·cos + rot(x)·sin) is fused into the attention
prologue when the HIP fused-attention path runs; this module provides the
embed generators + the standalone apply used on the eager path.
"""
import math
from typing import List, Optional, Tuple, Union

import torch
from torch import nn

from .helpers import to_2tuple


def pixel_freq_bands(
        num_bands: int,
        max_freq: float = 224.,
        linear_bands: bool = True,
        device: Optional[torch.device] = None,
):
    if linear_bands:
        bands = torch.linspace(1.0, max_freq / 2, num_bands, dtype=torch.float32, device=device)
    else:
        bands = 2 ** torch.linspace(0, math.log(max_freq, 2) - 1, num_bands, dtype=torch.float32, device=device)
    return bands * torch.pi


def freq_bands(
        num_bands: int,
        temperature: float = 10000.,
        step: int = 2,
        device: Optional[torch.device] = None,
) -> torch.Tensor:
    exp = torch.arange(0, num_bands, step, dtype=torch.int64, device=device).to(torch.float32) / num_bands
    bands = 1. / (temperature ** exp)
    return bands


def build_sincos2d_pos_embed(
        feat_shape: List[int],
        dim: int = 64,
        temperature: float = 10000.,
        reverse_coord: bool = False,
        interleave_sin_cos: bool = False,
        dtype: torch.dtype = torch.float32,
        device: Optional[torch.device] = None,
) -> torch.Tensor:
    """Fixed 2d sin-cos embed (reference `pos_embed_sincos.py:39`)."""
    assert dim % 4 == 0, 'Embed dimension must be divisible by 4 for sin-cos 2d position embedding'
    pos_dim = dim // 4
    bands = freq_bands(pos_dim, temperature=temperature, step=1, device=device)

    if reverse_coord:
        feat_shape = feat_shape[::-1]  # stack W, H instead of H, W
    grid = torch.stack(torch.meshgrid(
        [torch.arange(s, device=device, dtype=torch.int64).to(torch.float32) for s in feat_shape],
        indexing='ij'), dim=-1).unsqueeze(-1)
    pos2 = grid * bands
    pos2 = pos2.flatten(0, len(feat_shape) - 1)  # [N, 2, pos_dim]
    stack_dim = 2 if interleave_sin_cos else 1
    pos_emb = torch.stack([torch.sin(pos2), torch.cos(pos2)], dim=stack_dim).flatten(1)
    return pos_emb.to(dtype=dtype)


def build_fourier_pos_embed(
        feat_shape: List[int],
        bands: Optional[torch.Tensor] = None,
        num_bands: int = 64,
        max_res: int = 224,
        temperature: float = 10000.,
        linear_bands: bool = False,
        include_grid: bool = False,
        in_pixels: bool = True,
        ref_feat_shape: Optional[List[int]] = None,
        grid_offset: float = 0.,
        grid_indexing: str = 'ij',
        dtype: torch.dtype = torch.float32,
        device: Optional[torch.device] = None,
) -> List[torch.Tensor]:
    """Fourier (sin/cos over frequency bands) embed (reference `pos_embed_sincos.py:86`)."""
    if bands is None:
        if in_pixels:
            bands = pixel_freq_bands(num_bands, float(max_res), linear_bands=linear_bands, device=device)
        else:
            bands = freq_bands(num_bands, temperature=temperature, step=1, device=device)
    else:
        if device is None:
            device = bands.device
        if dtype is None:
            dtype = bands.dtype

    if in_pixels:
        t = [torch.linspace(-1., 1., steps=s, device=device, dtype=torch.float32) for s in feat_shape]
    else:
        t = [torch.arange(s, device=device, dtype=torch.int64).to(torch.float32) for s in feat_shape]
        if grid_offset:
            t = [x + grid_offset for x in t]

    if ref_feat_shape is not None:
        # eva's scheme for resizing rope embeddings (ref shape = pretrain)
        t = [x / f * r for x, f, r in zip(t, feat_shape, ref_feat_shape)]

    if grid_indexing == 'xy':
        t = list(reversed(t))
    grid = torch.stack(torch.meshgrid(t, indexing='ij'), dim=-1)
    if grid_indexing == 'xy':
        grid = grid.flip(-1)
    grid = grid.unsqueeze(-1)
    pos = grid * bands

    pos_sin, pos_cos = pos.sin().to(dtype=dtype), pos.cos().to(dtype=dtype)
    out = [grid, pos_sin, pos_cos] if include_grid else [pos_sin, pos_cos]
    return out


class FourierEmbed(nn.Module):
    def __init__(
            self,
            max_res: int = 224,
            num_bands: int = 64,
            concat_grid=True,
            keep_spatial=False,
    ):
        super().__init__()
        self.max_res = max_res
        self.num_bands = num_bands
        self.concat_grid = concat_grid
        self.keep_spatial = keep_spatial
        self.register_buffer(
            'bands', pixel_freq_bands(num_bands, max_res), persistent=False)

    def forward(self, x):
        B, C = x.shape[:2]
        feat_shape = x.shape[2:]
        emb = build_fourier_pos_embed(
            feat_shape, self.bands, include_grid=self.concat_grid,
            dtype=x.dtype, device=x.device)
        emb = torch.cat(emb, dim=-1)
        emb = emb.transpose(-1, -2).flatten(len(feat_shape))
        batch_expand = (B,) + (-1,) * (x.ndim - 1)
        if self.keep_spatial:
            x = torch.cat([x, emb.unsqueeze(0).expand(batch_expand).permute(0, 3, 1, 2)], dim=1)
        else:
            x = torch.cat([x.permute(0, 2, 3, 1), emb.unsqueeze(0).expand(batch_expand)], dim=-1)
            x = x.reshape(B, feat_shape.numel(), -1)
        return x


def rot(x):
    """rotate-half: (-x2, x1, -x4, x3, ...) — reference `pos_embed_sincos.py:228`."""
    return torch.stack([-x[..., 1::2], x[..., ::2]], -1).reshape(x.shape)


def rope_rotate_half(x: torch.Tensor) -> torch.Tensor:
    """NumPy/DeepSpeed-style half rotation: (-x[d/2:], x[:d/2])."""
    x1, x2 = x.chunk(2, dim=-1)
    return torch.cat([-x2, x1], dim=-1)


def apply_rot_embed(x: torch.Tensor, sin_emb, cos_emb):
    if sin_emb.ndim == 3:
        return x * cos_emb.unsqueeze(1).expand_as(x) + rot(x) * sin_emb.unsqueeze(1).expand_as(x)
    return x * cos_emb + rot(x) * sin_emb


def apply_rot_embed_list(x: List[torch.Tensor], sin_emb, cos_emb) -> List[torch.Tensor]:
    if isinstance(x, torch.Tensor):
        x = [x]
    return [t * cos_emb + rot(t) * sin_emb for t in x]


def apply_rot_embed_cat(x: torch.Tensor, emb) -> torch.Tensor:
    """Apply concatenated [sin, cos] embed: x·cos + rot(x)·sin (reference `:281`)."""
    sin_emb, cos_emb = emb.tensor_split(2, -1)
    if sin_emb.ndim == 3:
        return x * cos_emb.unsqueeze(1).expand_as(x) + rot(x) * sin_emb.unsqueeze(1).expand_as(x)
    return x * cos_emb + rot(x) * sin_emb


def apply_keep_indices_nlc(
        x: torch.Tensor,
        pos_embed: torch.Tensor,
        keep_indices: torch.Tensor,
        pos_embed_has_batch: bool = False,
) -> torch.Tensor:
    """Apply keep indices (from PatchDropout) to select position embeddings."""
    if pos_embed_has_batch:
        pos_embed = pos_embed.gather(
            1, keep_indices.unsqueeze(-1).expand(-1, -1, pos_embed.shape[-1]))
    else:
        pos_embed = pos_embed.unsqueeze(0).expand(x.shape[0], -1, -1)
        pos_embed = pos_embed.gather(
            1, keep_indices.unsqueeze(-1).expand(-1, -1, pos_embed.shape[-1]))
    return pos_embed


def build_rotary_pos_embed(
        feat_shape: List[int],
        bands: Optional[torch.Tensor] = None,
        dim: int = 64,
        max_res: int = 224,
        temperature: float = 10000.,
        linear_bands: bool = False,
        in_pixels: bool = True,
        ref_feat_shape: Optional[List[int]] = None,
        grid_offset: float = 0.,
        grid_indexing: str = 'ij',
        dtype: torch.dtype = torch.float32,
        device: Optional[torch.device] = None,
):
    """2D rotary embed sin/cos, each [N, dim] with per-axis bands repeated ×2."""
    sin_emb, cos_emb = build_fourier_pos_embed(
        feat_shape,
        bands=bands,
        num_bands=dim // 4,
        max_res=max_res,
        temperature=temperature,
        linear_bands=linear_bands,
        in_pixels=in_pixels,
        ref_feat_shape=ref_feat_shape,
        grid_offset=grid_offset,
        grid_indexing=grid_indexing,
        dtype=dtype,
        device=device,
    )
    num_spatial_dim = 1
    for x in feat_shape:
        num_spatial_dim *= x
    sin_emb = sin_emb.reshape(num_spatial_dim, -1).repeat_interleave(2, -1)
    cos_emb = cos_emb.reshape(num_spatial_dim, -1).repeat_interleave(2, -1)
    return sin_emb, cos_emb


class RotaryEmbedding(nn.Module):
    """Rotary embedding w/ separate sin/cos return (reference `pos_embed_sincos.py:393`)."""

    def __init__(
            self,
            dim,
            max_res=224,
            temperature=10000,
            in_pixels=True,
            linear_bands: bool = False,
            feat_shape: Optional[List[int]] = None,
            ref_feat_shape: Optional[List[int]] = None,
            grid_offset: float = 0.,
            grid_indexing: str = 'ij',
    ):
        super().__init__()
        self.dim = dim
        self.max_res = max_res
        self.temperature = temperature
        self.in_pixels = in_pixels
        self.linear_bands = linear_bands
        self.feat_shape = feat_shape
        self.ref_feat_shape = ref_feat_shape
        self.grid_offset = grid_offset
        self.grid_indexing = grid_indexing

        if feat_shape is None:
            # only cache bands
            if in_pixels:
                bands = pixel_freq_bands(dim // 4, float(max_res), linear_bands=linear_bands)
            else:
                bands = freq_bands(dim // 4, temperature=temperature, step=1)
            self.register_buffer('bands', bands, persistent=False)
            self.pos_embed_sin = None
            self.pos_embed_cos = None
        else:
            emb_sin, emb_cos = self._get_pos_embed_values(feat_shape)
            self.bands = None
            self.register_buffer('pos_embed_sin', emb_sin, persistent=False)
            self.register_buffer('pos_embed_cos', emb_cos, persistent=False)

    def _get_pos_embed_values(self, feat_shape: List[int]):
        return build_rotary_pos_embed(
            feat_shape=feat_shape,
            dim=self.dim,
            max_res=self.max_res,
            temperature=self.temperature,
            linear_bands=self.linear_bands,
            in_pixels=self.in_pixels,
            ref_feat_shape=self.ref_feat_shape,
            grid_offset=self.grid_offset,
            grid_indexing=self.grid_indexing,
        )

    def update_feat_shape(self, feat_shape: List[int]):
        if self.feat_shape is not None and feat_shape != self.feat_shape:
            assert self.pos_embed_sin is not None
            emb_sin, emb_cos = self._get_pos_embed_values(feat_shape)
            self.pos_embed_sin = emb_sin.to(self.pos_embed_sin.device, self.pos_embed_sin.dtype)
            self.pos_embed_cos = emb_cos.to(self.pos_embed_cos.device, self.pos_embed_cos.dtype)
            self.feat_shape = feat_shape

    def get_embed(self, shape: Optional[List[int]] = None):
        if shape is not None and self.bands is not None:
            return build_rotary_pos_embed(
                shape,
                self.bands,
                in_pixels=self.in_pixels,
                ref_feat_shape=self.ref_feat_shape,
                grid_offset=self.grid_offset,
                grid_indexing=self.grid_indexing,
            )
        elif self.pos_embed_sin is not None and self.pos_embed_cos is not None:
            return self.pos_embed_sin, self.pos_embed_cos
        raise AssertionError('get_embed() requires pre-computed pos embed or valid shape w/ pre-computed bands')

    def forward(self, x):
        sin_emb, cos_emb = self.get_embed(x.shape[2:])
        return apply_rot_embed(x, sin_emb, cos_emb)


class RotaryEmbeddingCat(nn.Module):
    """Rotary embedding w/ concatenated [sin, cos] (reference `pos_embed_sincos.py:534`).

    The form consumed by EVA02 / `apply_rot_embed_cat`.  `get_batch_embeds`
    supports NaFlex keep-indices selection (reference `:662`).
    """

    def __init__(
            self,
            dim,
            max_res=224,
            temperature=10000,
            in_pixels=True,
            linear_bands: bool = False,
            feat_shape: Optional[List[int]] = None,
            ref_feat_shape: Optional[List[int]] = None,
            grid_offset: float = 0.,
            grid_indexing: str = 'ij',
    ):
        super().__init__()
        self.dim = dim
        self.max_res = max_res
        self.temperature = temperature
        self.in_pixels = in_pixels
        self.linear_bands = linear_bands
        self.feat_shape = feat_shape
        self.ref_feat_shape = ref_feat_shape
        self.grid_offset = grid_offset
        self.grid_indexing = grid_indexing

        if feat_shape is None:
            if in_pixels:
                bands = pixel_freq_bands(dim // 4, float(max_res), linear_bands=linear_bands)
            else:
                bands = freq_bands(dim // 4, temperature=temperature, step=1)
            self.register_buffer('bands', bands, persistent=False)
            self.pos_embed = None
        else:
            self.bands = None
            self.register_buffer('pos_embed', self._get_pos_embed_values(feat_shape), persistent=False)

    def _get_pos_embed_values(self, feat_shape: List[int]):
        embeds = build_rotary_pos_embed(
            feat_shape=feat_shape,
            dim=self.dim,
            max_res=self.max_res,
            temperature=self.temperature,
            linear_bands=self.linear_bands,
            in_pixels=self.in_pixels,
            ref_feat_shape=self.ref_feat_shape,
            grid_offset=self.grid_offset,
            grid_indexing=self.grid_indexing,
        )
        return torch.cat(embeds, -1)

    def update_feat_shape(self, feat_shape: List[int]):
        if self.feat_shape is not None and feat_shape != self.feat_shape:
            assert self.pos_embed is not None
            self.pos_embed = self._get_pos_embed_values(feat_shape).to(
                device=self.pos_embed.device, dtype=self.pos_embed.dtype)
            self.feat_shape = feat_shape

    def get_embed(self, shape: Optional[List[int]] = None):
        if shape is not None and self.bands is not None:
            embeds = build_rotary_pos_embed(
                shape,
                self.bands,
                in_pixels=self.in_pixels,
                ref_feat_shape=self.ref_feat_shape,
                grid_offset=self.grid_offset,
                grid_indexing=self.grid_indexing,
            )
            return torch.cat(embeds, -1)
        elif self.pos_embed is not None:
            return self.pos_embed
        raise AssertionError('get_embed() requires pre-computed pos embed or valid shape w/ pre-computed bands')

    def get_batch_embeds(
            self,
            shapes: List[Tuple[int, int]],
            seq_len: Optional[int] = None,
    ):
        """Generate ROPE embeddings for multiple grid shapes efficiently (NaFlex)."""
        if self.bands is None:
            raise RuntimeError('get_batch_embeds requires bands (feat_shape=None mode)')
        max_h = max(h for h, w in shapes)
        max_w = max(w for h, w in shapes)
        sin_emb, cos_emb = build_rotary_pos_embed(
            feat_shape=(max_h, max_w),
            bands=self.bands,
            in_pixels=self.in_pixels,
            ref_feat_shape=self.ref_feat_shape,
            grid_offset=self.grid_offset,
            grid_indexing=self.grid_indexing,
        )
        rope_embeds = torch.cat([sin_emb, cos_emb], -1).reshape(max_h, max_w, -1)
        if seq_len is not None:
            flat_embeds = []
            for h, w in shapes:
                emb = rope_embeds[:h, :w].reshape(h * w, -1)
                pad_len = seq_len - emb.shape[0]
                if pad_len > 0:
                    emb = torch.cat([emb, emb.new_zeros(pad_len, emb.shape[-1])], 0)
                flat_embeds.append(emb)
            return torch.stack(flat_embeds)
        return [rope_embeds[:h, :w].reshape(h * w, -1) for h, w in shapes]

    def forward(self, x):
        pos_embed = self.get_embed(x.shape[2:])
        return apply_rot_embed_cat(x, pos_embed)


def create_rope_embed(
        rope_type: str = 'cat',
        dim: int = 768,
        num_heads: int = 12,
        **kwargs,
):
    """RoPE factory (reference `pos_embed_sincos.py:1315`).

    ``dim`` is the TOTAL embedding dim; per-head dim is derived via num_heads.
    """
    if rope_type in ('', 'cat', 'regular'):
        if kwargs.pop('rotate_half', False):
            raise NotImplementedError(
                'rotate_half RoPE layout is not implemented for the cat variant; '
                'interleaved rotation would give wrong outputs silently')
        return RotaryEmbeddingCat(dim // num_heads, **kwargs)
    if rope_type == 'base':
        if kwargs.pop('rotate_half', False):
            raise NotImplementedError('rotate_half RoPE layout is not implemented')
        return RotaryEmbedding(dim // num_heads, **kwargs)
    raise ValueError(f'Unknown rope type {rope_type}')
