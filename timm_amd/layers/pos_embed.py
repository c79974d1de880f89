"""Learned absolute position-embedding resampling (variable input sizes).

Behavioral parity: /root/reference/timm/layers/pos_embed.py:19 (prefix-token
split, fp32 bicubic+antialias interpolation, dtype round-trip).
"""
import logging
import math
from typing import List, Optional

import torch
import torch.nn.functional as F

_logger = logging.getLogger(__name__)

__all__ = ['resample_abs_pos_embed', 'resample_abs_pos_embed_nhwc']


def _interp_grid(flat: torch.Tensor, old_size, new_size, interpolation, antialias):
    """[1, H*W, C] -> [1, H'*W', C] via NCHW interpolation in fp32."""
    dtype = flat.dtype
    dim = flat.shape[-1]
    grid = flat.float().reshape(1, old_size[0], old_size[1], dim).permute(0, 3, 1, 2)
    grid = F.interpolate(grid, size=new_size, mode=interpolation, antialias=antialias)
    return grid.permute(0, 2, 3, 1).reshape(1, -1, dim).to(dtype)


def resample_abs_pos_embed(
        posemb: torch.Tensor,
        new_size: List[int],
        old_size: Optional[List[int]] = None,
        num_prefix_tokens: int = 1,
        interpolation: str = 'bicubic',
        antialias: bool = True,
        verbose: bool = False,
):
    """Resample a [1, prefix + H*W, C] pos-embed table to a new grid.

    Prefix (cls/reg) tokens pass through untouched; the grid is assumed
    square when old_size is not given.
    """
    total = posemb.shape[1]
    if total == new_size[0] * new_size[1] + num_prefix_tokens and new_size[0] == new_size[1]:
        return posemb
    if old_size is None:
        side = int(math.sqrt(total - num_prefix_tokens))
        old_size = side, side

    prefix = posemb[:, :num_prefix_tokens] if num_prefix_tokens else None
    grid = posemb[:, num_prefix_tokens:] if num_prefix_tokens else posemb
    grid = _interp_grid(grid, old_size, new_size, interpolation, antialias)
    if prefix is not None:
        grid = torch.cat([prefix, grid], dim=1)

    if not torch.jit.is_scripting() and verbose:
        _logger.info(f'Resized position embedding: {old_size} to {new_size}.')
    return grid


def resample_abs_pos_embed_nhwc(
        posemb: torch.Tensor,
        new_size: List[int],
        interpolation: str = 'bicubic',
        antialias: bool = True,
        verbose: bool = False,
):
    """Resample an NHWC-laid-out pos-embed [..., H, W, C] to a new grid."""
    if tuple(new_size) == (posemb.shape[-3], posemb.shape[-2]):
        return posemb
    dtype = posemb.dtype
    grid = posemb.float().reshape(
        1, posemb.shape[-3], posemb.shape[-2], posemb.shape[-1]).permute(0, 3, 1, 2)
    grid = F.interpolate(grid, size=new_size, mode=interpolation, antialias=antialias)
    out = grid.permute(0, 2, 3, 1).to(dtype)
    if not torch.jit.is_scripting() and verbose:
        _logger.info(f'Resized position embedding to {new_size}.')
    return out
