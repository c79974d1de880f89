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


def apply_rot_embed_cat(x: torch.Tensor, emb, half: bool = False) -> torch.Tensor:
    """Apply concatenated [sin, cos] embed: x·cos + rotate(x)·sin (reference `:281`).

    ``half`` selects the half-rotation layout ([-x[d/2:], x[:d/2]], DINOv3
    checkpoints) instead of the interleaved pair rotation.
    """
    sin_emb, cos_emb = emb.tensor_split(2, -1)
    rotate = rope_rotate_half if half else rot
    # plain broadcasting: callers align batch/head dims (e.g. NaFlex batched
    # embeds arrive as [B, 1, N, D]; mixed-mode as [num_heads, N, D])
    return x * cos_emb + rotate(x) * sin_emb


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




# ---------------------------------------------------------------------------
# mixed (learnable, per-depth) RoPE — rope-vit arxiv 2403.13298
# ---------------------------------------------------------------------------

def init_random_2d_freqs(
        head_dim: int,
        depth: int,
        num_heads: int,
        temperature: float = 10.0,
        rotate: bool = True,
        *,
        device=None,
        dtype=torch.float32,
) -> torch.Tensor:
    """Per-(depth, head) randomly-rotated 2D frequency pairs.

    Returns [2, depth, num_heads, head_dim//2] (x-freqs stacked over y-freqs).
    """
    mag = 1.0 / (temperature ** (
        torch.arange(0, head_dim, 4, device=device, dtype=dtype) / head_dim))
    mag = mag.view(1, 1, -1)
    if rotate:
        angles = torch.rand(depth, num_heads, 1, device=device, dtype=dtype) * 2 * torch.pi
    else:
        angles = torch.zeros(depth, num_heads, 1, device=device, dtype=dtype)
    fx = torch.cat([mag * torch.cos(angles), mag * torch.cos(angles + torch.pi / 2)], dim=-1)
    fy = torch.cat([mag * torch.sin(angles), mag * torch.sin(angles + torch.pi / 2)], dim=-1)
    return torch.stack([fx, fy], dim=0)


def get_mixed_grid(
        shape: List[int],
        grid_indexing: str = 'ij',
        device: Optional[torch.device] = None,
        dtype: torch.dtype = torch.float32,
) -> Tuple[torch.Tensor, torch.Tensor]:
    if grid_indexing == 'xy':
        shape = [shape[1], shape[0]]
    x_pos, y_pos = torch.meshgrid(
        torch.arange(shape[0], device=device, dtype=torch.float32),
        torch.arange(shape[1], device=device, dtype=torch.float32),
        indexing=grid_indexing,
    )
    return x_pos.to(dtype).flatten(), y_pos.to(dtype).flatten()


def get_mixed_freqs(
        freqs: torch.Tensor,
        t_x: torch.Tensor,
        t_y: torch.Tensor,
) -> torch.Tensor:
    """Project grid coords through the learnable frequency pairs.

    Returns [depth, num_heads, N, head_dim] with sin||cos concatenated.
    """
    dtype = freqs.dtype
    freqs = freqs.float()
    fx = t_x.unsqueeze(-1) @ freqs[0].unsqueeze(-2)
    fy = t_y.unsqueeze(-1) @ freqs[1].unsqueeze(-2)
    combined = fx + fy
    sin_emb = torch.sin(combined).repeat_interleave(2, -1)
    cos_emb = torch.cos(combined).repeat_interleave(2, -1)
    return torch.cat([sin_emb, cos_emb], dim=-1).to(dtype)


class RotaryEmbeddingMixed(nn.Module):
    """Depth-dependent learnable RoPE frequencies (rope-vit 'mixed' mode).

    One learnable (x, y) frequency pair per (block, head); ``get_embed``
    yields [depth, num_heads, N, head_dim] so the model indexes per block.
    Reference: timm/layers/pos_embed_sincos.py:873.
    """

    def __init__(
            self,
            dim: int,
            depth: int,
            num_heads: int,
            temperature: float = 10.0,
            feat_shape: Optional[List[int]] = None,
            grid_indexing: str = 'xy',
    ):
        super().__init__()
        self.dim = dim
        self.depth = depth
        self.num_heads = num_heads
        self.temperature = temperature
        self.feat_shape = feat_shape
        self.grid_indexing = grid_indexing

        head_dim = dim // num_heads
        assert head_dim % 4 == 0, f'head_dim must be divisible by 4, got {head_dim}'
        self.freqs = nn.Parameter(init_random_2d_freqs(
            head_dim, depth, num_heads, temperature=temperature, rotate=True))

        if feat_shape is not None:
            n = 1
            for v in feat_shape:
                n *= v
            self.register_buffer('t_x', torch.empty(n), persistent=False)
            self.register_buffer('t_y', torch.empty(n), persistent=False)
            self._init_buffers()
        else:
            self.t_x = self.t_y = None

    def _init_buffers(self):
        if self.feat_shape is not None:
            t_x, t_y = get_mixed_grid(
                self.feat_shape, grid_indexing=self.grid_indexing, device=self.freqs.device)
            self.t_x.copy_(t_x)
            self.t_y.copy_(t_y)

    def reset_parameters(self):
        self._init_buffers()

    def init_non_persistent_buffers(self):
        self._init_buffers()

    def update_feat_shape(self, feat_shape: Optional[List[int]]):
        if self.feat_shape is not None and feat_shape != self.feat_shape:
            t_x, t_y = get_mixed_grid(
                feat_shape, grid_indexing=self.grid_indexing, device=self.freqs.device)
            self.t_x = t_x.to(self.t_x.device, self.t_x.dtype)
            self.t_y = t_y.to(self.t_y.device, self.t_y.dtype)
            self.feat_shape = feat_shape

    def get_embed(self, shape: Optional[List[int]] = None) -> torch.Tensor:
        if shape is not None:
            t_x, t_y = get_mixed_grid(
                shape, grid_indexing=self.grid_indexing, device=self.freqs.device)
        else:
            assert self.t_x is not None and self.t_y is not None, \
                'get_embed() requires pre-computed t_x/t_y or valid shape'
            t_x, t_y = self.t_x, self.t_y
        return get_mixed_freqs(self.freqs, t_x, t_y)

    def get_batch_embeds(self, shapes: List[Tuple[int, int]], seq_len: Optional[int] = None):
        """Per-shape flattened embeds (NaFlex); pads to seq_len when given."""
        max_h = max(h for h, _ in shapes)
        max_w = max(w for _, w in shapes)
        full = self.get_embed((max_h, max_w))  # [depth, heads, H*W, dim]
        full = full.reshape(self.depth, self.num_heads, max_h, max_w, -1)
        out = []
        for h, w in shapes:
            emb = full[:, :, :h, :w].reshape(self.depth, self.num_heads, h * w, -1)
            if seq_len is not None and emb.shape[2] < seq_len:
                pad = emb.new_zeros(self.depth, self.num_heads, seq_len - emb.shape[2], emb.shape[-1])
                emb = torch.cat([emb, pad], dim=2)
            out.append(emb)
        return torch.stack(out, dim=1) if seq_len is not None else out

    def forward(self, x):
        return apply_rot_embed_cat(x, self.get_embed(x.shape[2:]))

    def no_weight_decay(self):
        return {'freqs'}


# ---------------------------------------------------------------------------
# DINOv3 RoPE — normalized [-1,1] coords, period schedule, train-time augs
# ---------------------------------------------------------------------------

def make_coords_dinov3(
        height: int,
        width: int,
        normalize_coords: str = 'separate',
        grid_indexing: str = 'ij',
        grid_offset: float = 0.,
        device: torch.device = 'cpu',
        dtype: torch.dtype = torch.float32,
) -> torch.Tensor:
    """0.5-centered patch coords normalized to [-1, 1]; returns [HW, 2]."""
    ch = torch.arange(0.5, height, device=device, dtype=torch.float32) + grid_offset
    cw = torch.arange(0.5, width, device=device, dtype=torch.float32) + grid_offset
    if normalize_coords == 'max':
        h_denom = w_denom = float(max(height, width))
    elif normalize_coords == 'min':
        h_denom = w_denom = float(min(height, width))
    elif normalize_coords == 'separate':
        h_denom, w_denom = float(height), float(width)
    else:
        raise ValueError(f'Unknown normalize_coords: {normalize_coords}')
    ch = 2.0 * ch / h_denom - 1.0
    cw = 2.0 * cw / w_denom - 1.0
    if grid_indexing == 'xy':
        gw, gh = torch.meshgrid(cw, ch, indexing='ij')
        coords = torch.stack([gh, gw], dim=-1)
    else:
        gh, gw = torch.meshgrid(ch, cw, indexing='ij')
        coords = torch.stack([gh, gw], dim=-1)
    return coords.flatten(0, 1).to(dtype)


class RotaryEmbeddingDinoV3(nn.Module):
    """DINOv3-numerics RoPE: normalized coords, min/max-period (or
    temperature) schedule, optional shift/jitter/rescale train augs, half or
    interleaved rotation layout.  Reference: pos_embed_sincos.py:1107."""

    def __init__(
            self,
            dim: int,
            temperature: Optional[float] = 100.0,
            min_period: Optional[float] = None,
            max_period: Optional[float] = None,
            feat_shape: Optional[List[int]] = None,
            normalize_coords: str = 'separate',
            grid_offset: float = 0.0,
            grid_indexing: str = 'ij',
            rotate_half: bool = True,
            shift_coords: Optional[float] = None,
            jitter_coords: Optional[float] = None,
            rescale_coords: Optional[float] = None,
    ):
        super().__init__()
        self.dim = dim
        self.rotate_half = rotate_half
        self.temperature = float(temperature) if temperature is not None else None
        self.min_period = min_period
        self.max_period = max_period
        self.normalize_coords = normalize_coords
        self.shift_coords = shift_coords
        self.jitter_coords = jitter_coords
        self.rescale_coords = rescale_coords
        self.aug_active = any(
            a is not None for a in (shift_coords, jitter_coords, rescale_coords))
        self.feat_shape = feat_shape
        self.grid_offset = grid_offset
        self.grid_indexing = grid_indexing

        self.register_buffer('periods', torch.empty(dim // 4), persistent=False)
        if feat_shape is not None:
            n = feat_shape[0] * feat_shape[1]
            self.register_buffer('pos_embed_cached', torch.empty(n, dim * 2), persistent=False)
        else:
            self.pos_embed_cached = None
        self.reset_parameters()

    def _compute_periods(self) -> torch.Tensor:
        n = self.dim // 4
        if self.min_period is not None and self.max_period is not None:
            exp = torch.linspace(0, 1, n)
            return self.min_period * ((self.max_period / self.min_period) ** exp)
        if self.temperature is None:
            raise ValueError('Provide either min/max periods or `temperature`.')
        exp = 2.0 * torch.arange(n, dtype=torch.float32) / (self.dim // 2)
        return self.temperature ** exp

    def reset_parameters(self):
        self._init_buffers()

    def _init_buffers(self):
        self.periods.copy_(self._compute_periods())
        if self.feat_shape is not None and self.pos_embed_cached is not None:
            self.pos_embed_cached.copy_(self._create_embed(self.feat_shape, no_aug=True))

    def init_non_persistent_buffers(self):
        self._init_buffers()

    def _augment(self, coords: torch.Tensor) -> torch.Tensor:
        if not self.training or not self.aug_active:
            return coords
        device, dtype = coords.device, coords.dtype
        if self.shift_coords is not None:
            s = float(self.shift_coords)
            coords = coords + torch.empty(2, device=device, dtype=dtype).uniform_(-s, s)[None, :]
        if self.jitter_coords is not None:
            j = math.log(float(self.jitter_coords))
            coords = coords * torch.empty(2, device=device, dtype=dtype).uniform_(-j, j).exp()[None, :]
        if self.rescale_coords is not None:
            r = math.log(float(self.rescale_coords))
            coords = coords * torch.empty(1, device=device, dtype=dtype).uniform_(-r, r).exp()
        return coords

    def _angles_to_sincos(self, coords: torch.Tensor):
        coords = coords[:, :, None].to(device=self.periods.device, dtype=self.periods.dtype)
        angles = (2 * math.pi * coords / self.periods[None, None, :]).flatten(1)
        if self.rotate_half:
            angles = angles.tile(2)
        else:
            angles = angles.repeat_interleave(2, dim=-1)
        return torch.sin(angles), torch.cos(angles)

    def _create_embed(self, feat_shape: List[int], no_aug: bool = False) -> torch.Tensor:
        coords = make_coords_dinov3(
            feat_shape[0], feat_shape[1],
            normalize_coords=self.normalize_coords,
            grid_indexing=self.grid_indexing,
            grid_offset=self.grid_offset,
        )
        if not no_aug:
            coords = self._augment(coords)
        sin, cos = self._angles_to_sincos(coords)
        return torch.cat([sin, cos], dim=-1)

    def update_feat_shape(self, feat_shape: List[int]):
        if self.feat_shape is not None and feat_shape != self.feat_shape:
            emb = self._create_embed(feat_shape, no_aug=True)
            self.register_buffer('pos_embed_cached', emb, persistent=False)
            self.feat_shape = feat_shape

    def get_embed(self, shape: Optional[List[int]] = None) -> torch.Tensor:
        if shape is not None:
            return self._create_embed(shape)
        if self.pos_embed_cached is not None and not (self.training and self.aug_active):
            return self.pos_embed_cached
        assert self.feat_shape is not None, 'feature shape must be cached on create'
        return self._create_embed(self.feat_shape)

    def get_batch_embeds(self, shapes: List[Tuple[int, int]], seq_len: Optional[int] = None):
        embeds = [self._create_embed((h, w)) for h, w in shapes]
        if seq_len is None:
            return embeds
        out = []
        for emb in embeds:
            if emb.shape[0] < seq_len:
                emb = torch.cat(
                    [emb, emb.new_zeros(seq_len - emb.shape[0], emb.shape[-1])], 0)
            out.append(emb[:seq_len])
        return torch.stack(out)

    def forward(self, x):
        return apply_rot_embed_cat(x, self.get_embed(x.shape[2:]), half=self.rotate_half)



def create_rope_embed(
        rope_type: str = 'cat',
        dim: int = 768,
        num_heads: int = 12,
        **kwargs,
):
    """RoPE factory (reference `pos_embed_sincos.py:1315`).

    ``dim`` is the TOTAL embedding dim; per-head dim derived via num_heads
    except for 'mixed' (per-depth learnable freqs take the full dim).
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
    if rope_type == 'mixed':
        kwargs.pop('rotate_half', None)  # mixed mode is always interleaved
        kwargs.pop('in_pixels', None)
        kwargs.pop('grid_offset', None)
        kwargs.pop('ref_feat_shape', None)
        return RotaryEmbeddingMixed(dim, num_heads=num_heads, **kwargs)
    if rope_type == 'dinov3':
        kwargs.pop('in_pixels', None)
        kwargs.pop('ref_feat_shape', None)
        return RotaryEmbeddingDinoV3(dim // num_heads, **kwargs)
    raise ValueError(f'Unknown rope type {rope_type}')
