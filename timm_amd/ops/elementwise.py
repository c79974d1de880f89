"""Fused elementwise epilogues.

 * bias_act: y = act(x + bias) — the GEMM bias+GELU/SiLU epilogue
   (replaces `nn.Linear` bias + activation pair, reference `timm/layers/mlp.py:40-44`).
 * residual_scale_add: out = x + drop_path(gamma * y) — the residual epilogue
   with LayerScale + per-sample DropPath folded in
   (reference `vision_transformer.py:212-213`, `timm/layers/drop.py:158`).

Device path: single HIP kernel each (HBM-bound, vectorized bf16x8).
CPU path: eager composition (numerics oracle).
"""
from typing import Optional

import torch
import torch.nn.functional as F

from . import _load_extension

_ACT_IDS = {'gelu': 0, 'gelu_tanh': 1, 'silu': 2, 'relu': 3, 'identity': 4, 'quick_gelu': 5}


class _BiasActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias, act_id):
        ext = _load_extension()
        x = x.contiguous()
        y = ext.bias_act_fwd(x, bias, act_id)
        ctx.save_for_backward(x, bias)
        ctx.act_id = act_id
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _load_extension()
        x, bias = ctx.saved_tensors
        dx, db = ext.bias_act_bwd(dy.contiguous(), x, bias, ctx.act_id)
        return dx, db, None


def _act_eager(x, act: str):
    if act == 'gelu':
        return F.gelu(x)
    if act == 'gelu_tanh':
        return F.gelu(x, approximate='tanh')
    if act == 'silu':
        return F.silu(x)
    if act == 'relu':
        return F.relu(x)
    if act == 'quick_gelu':
        return x * torch.sigmoid(1.702 * x)
    return x


def bias_act(x: torch.Tensor, bias: Optional[torch.Tensor], act: str = 'gelu') -> torch.Tensor:
    """act(x + bias) over last dim; act in {gelu, gelu_tanh, silu, relu, identity, quick_gelu}."""
    if x.is_cuda and _load_extension() is not None and bias is not None and act in _ACT_IDS \
            and x.dtype in (torch.bfloat16, torch.float16) and bias.dtype == x.dtype:
        return _BiasActFn.apply(x, bias, _ACT_IDS[act])
    if x.is_cuda:
        from . import use_hip
        use_hip(x)
    if bias is not None:
        x = x + bias
    return _act_eager(x, act)


class _ResidualScaleAddFn(torch.autograd.Function):
    """out = x + y * gamma * keep_mask  (gamma: [C] or None; keep_mask: [B] or None)"""

    @staticmethod
    def forward(ctx, x, y, gamma, keep_mask):
        ext = _load_extension()
        out = ext.residual_scale_add_fwd(x.contiguous(), y.contiguous(), gamma, keep_mask)
        ctx.save_for_backward(y, gamma, keep_mask)
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = _load_extension()
        y, gamma, keep_mask = ctx.saved_tensors
        dout = dout.contiguous()
        dx = dout
        dy, dgamma = ext.residual_scale_add_bwd(
            dout, y if gamma is not None else None, gamma, keep_mask)
        return dx, dy, dgamma, None


def residual_scale_add(
        x: torch.Tensor,
        y: torch.Tensor,
        gamma: Optional[torch.Tensor] = None,
        drop_prob: float = 0.,
        training: bool = False,
        scale_by_keep: bool = True,
) -> torch.Tensor:
    """x + drop_path(y * gamma). Fuses residual-add + LayerScale + DropPath."""
    keep_mask = None
    if drop_prob > 0. and training:
        keep_prob = 1. - drop_prob
        keep_mask = torch.empty(x.shape[0], device=x.device, dtype=torch.float32).bernoulli_(keep_prob)
        if scale_by_keep and keep_prob > 0.:
            keep_mask = keep_mask / keep_prob
    if x.is_cuda and _load_extension() is not None and x.is_contiguous() and y.is_contiguous() \
            and x.dtype in (torch.bfloat16, torch.float16) and y.dtype == x.dtype \
            and (gamma is None or (gamma.ndim == 1 and gamma.dtype == x.dtype)):
        return _ResidualScaleAddFn.apply(x, y, gamma, keep_mask)
    if x.is_cuda:
        from . import use_hip
        use_hip(x)
    if gamma is not None:
        y = y * gamma
    if keep_mask is not None:
        y = y * keep_mask.view(-1, *([1] * (x.ndim - 1))).to(y.dtype)
    return x + y
