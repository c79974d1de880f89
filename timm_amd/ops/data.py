"""Data-path ops (gfx950 HIP kernels, `csrc/data_ops.hip`).

* `u8_normalize` — fused uint8 -> normalized bf16/fp16/fp32 conversion used by
  the loader prefetcher. Replaces the reference's
  `.to(dtype).sub_(mean).div_(std)` three-kernel chain
  (`timm/data/loader.py:116`) with one bandwidth-bound pass.
* `masked_global_pool` — masked avg/max pooling over valid NaFlex tokens
  (`timm/models/naflexvit.py:1065`), fused mask+reduce with an autograd
  backward that never materialises the [B, N, C] mask-broadcast product.
"""
from typing import Optional

import torch

from . import require_ext, use_hip

__all__ = ['u8_normalize', 'masked_global_pool']


def u8_normalize(
        x: torch.Tensor,
        mean: torch.Tensor,
        std: torch.Tensor,
        dtype: torch.dtype = torch.float32,
) -> torch.Tensor:
    """(x - mean[c]) / std[c] for NCHW uint8 input; mean/std are per-channel
    (any broadcastable shape, on the 0-255 scale)."""
    if x.is_cuda and x.dtype == torch.uint8 and x.dim() == 4 and x.numel() % 4 == 0 and use_hip(x):
        ext = require_ext()
        c = x.size(1)
        mean_c = mean.reshape(-1).float()
        std_c = std.reshape(-1).float()
        if mean_c.numel() == 1:
            mean_c = mean_c.expand(c).contiguous()
        if std_c.numel() == 1:
            std_c = std_c.expand(c).contiguous()
        return ext.u8_normalize(x, mean_c, 1.0 / std_c, dtype)
    return x.to(dtype).sub_(mean.to(dtype)).div_(std.to(dtype))


class _MaskedPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, valid, is_max):
        ext = require_ext()
        out, aux = ext.masked_pool_fwd(x, valid, is_max)
        ctx.save_for_backward(valid, aux)
        ctx.n_tokens = x.shape[1]
        ctx.is_max = is_max
        return out

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        valid, aux = ctx.saved_tensors
        dx = ext.masked_pool_bwd(dy.contiguous(), valid, aux, ctx.n_tokens, ctx.is_max)
        return dx, None, None


def masked_global_pool(
        x: torch.Tensor,
        valid: torch.Tensor,
        pool_type: str = 'avg',
) -> torch.Tensor:
    """Pool [B, N, C] over the tokens where valid[B, N] is True."""
    assert pool_type in ('avg', 'max', 'avgmax')
    if use_hip(x):
        if pool_type == 'avg':
            return _MaskedPoolFn.apply(x.contiguous(), valid, False)
        if pool_type == 'max':
            return _MaskedPoolFn.apply(x.contiguous(), valid, True)
        return 0.5 * (_MaskedPoolFn.apply(x.contiguous(), valid, False)
                      + _MaskedPoolFn.apply(x.contiguous(), valid, True))
    # CPU reference
    vm = valid.to(x.dtype)
    denom = vm.sum(dim=1, keepdim=True).clamp(min=1)
    avg = (x * vm.unsqueeze(-1)).sum(dim=1) / denom
    if pool_type == 'avg':
        return avg
    masked = x.masked_fill(~valid.bool().unsqueeze(-1), torch.finfo(x.dtype).min)
    mx = masked.amax(dim=1)
    if pool_type == 'max':
        return mx
    return 0.5 * (avg + mx)
