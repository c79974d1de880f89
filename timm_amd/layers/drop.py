"""DropPath (stochastic depth) + DropBlock (reference `timm/layers/drop.py:102-193`).

On device, the residual-add + per-sample DropPath scale is fused into the HIP
residual epilogue kernel (see `timm_amd/ops`); this module provides the module
API + a standalone composable implementation.
"""
from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


def drop_path(x, drop_prob: float = 0., training: bool = False, scale_by_keep: bool = True):
    """Drop paths (Stochastic Depth) per sample, applied in the main path of residual blocks."""
    if drop_prob == 0. or not training:
        return x
    keep_prob = 1 - drop_prob
    shape = (x.shape[0],) + (1,) * (x.ndim - 1)
    random_tensor = x.new_empty(shape).bernoulli_(keep_prob)
    if keep_prob > 0.0 and scale_by_keep:
        random_tensor.div_(keep_prob)
    return x * random_tensor


class DropPath(nn.Module):
    def __init__(self, drop_prob: float = 0., scale_by_keep: bool = True):
        super().__init__()
        self.drop_prob = drop_prob
        self.scale_by_keep = scale_by_keep

    def forward(self, x):
        return drop_path(x, self.drop_prob, self.training, self.scale_by_keep)

    def extra_repr(self):
        return f'drop_prob={round(self.drop_prob, 3):0.3f}'


def calculate_drop_path_rates(
        drop_path_rate: float,
        depths,
        stagewise: bool = False,
) -> List:
    """Generate drop path rates, linearly increasing through depth.

    Reference `timm/layers/drop.py:193`.  `depths` may be an int (returns a flat
    list) or a per-stage tuple (with `stagewise`, rate constant within a stage).
    """
    if isinstance(depths, int):
        depths = [depths]
        squeeze = True
    else:
        squeeze = False
    total = sum(depths)
    if stagewise:
        dpr_stages = torch.linspace(0, drop_path_rate, len(depths)).tolist()
        out = [[dpr_stages[i]] * d for i, d in enumerate(depths)]
    else:
        rates = torch.linspace(0, drop_path_rate, total).tolist()
        out = []
        idx = 0
        for d in depths:
            out.append(rates[idx:idx + d])
            idx += d
    if squeeze:
        return out[0]
    return out


def drop_block_2d(
        x, drop_prob: float = 0.1, block_size: int = 7, gamma_scale: float = 1.0,
        with_noise: bool = False, inplace: bool = False, batchwise: bool = False):
    B, C, H, W = x.shape
    total_size = W * H
    clipped_block_size = min(block_size, min(W, H))
    gamma = gamma_scale * drop_prob * total_size / clipped_block_size ** 2 / (
            (W - block_size + 1) * (H - block_size + 1))

    if batchwise:
        block_mask = torch.rand((1, C, H, W), dtype=x.dtype, device=x.device) < gamma
    else:
        block_mask = torch.rand_like(x) < gamma
    block_mask = F.max_pool2d(
        block_mask.to(x.dtype), kernel_size=clipped_block_size, stride=1, padding=clipped_block_size // 2)

    if with_noise:
        normal_noise = torch.randn((1, C, H, W), dtype=x.dtype, device=x.device) if batchwise else torch.randn_like(x)
        if inplace:
            x.mul_(1. - block_mask).add_(normal_noise * block_mask)
        else:
            x = x * (1. - block_mask) + normal_noise * block_mask
    else:
        block_mask = 1 - block_mask
        normalize_scale = (block_mask.numel() / block_mask.to(dtype=torch.float32).sum().add(1e-7)).to(x.dtype)
        if inplace:
            x.mul_(block_mask * normalize_scale)
        else:
            x = x * block_mask * normalize_scale
    return x


class DropBlock2d(nn.Module):
    def __init__(
            self, drop_prob: float = 0.1, block_size: int = 7, gamma_scale: float = 1.0,
            with_noise: bool = False, inplace: bool = False, batchwise: bool = False, fast: bool = True):
        super().__init__()
        self.drop_prob = drop_prob
        self.gamma_scale = gamma_scale
        self.block_size = block_size
        self.with_noise = with_noise
        self.inplace = inplace
        self.batchwise = batchwise
        self.fast = fast

    def forward(self, x):
        if not self.training or not self.drop_prob:
            return x
        return drop_block_2d(
            x, self.drop_prob, self.block_size, self.gamma_scale, self.with_noise, self.inplace, self.batchwise)
