"""Fused cross-entropy loss (gfx950 HIP kernel, `csrc/ce_loss.hip`).

Replaces the reference's log_softmax + gather / soft-target chain
(`timm/loss/cross_entropy.py:1-36`): forward computes per-row loss and
log-sum-exp in one HBM pass over the logits; backward recomputes softmax from
the saved LSE and writes dlogits directly — the [B, C] softmax intermediate
never exists in memory.

CPU tensors fall back to the plain PyTorch formulation (which is also the
numerics reference for the GPU tests).
"""
from typing import Optional

import torch
import torch.nn.functional as F

from . import require_ext, use_hip

__all__ = ['fused_cross_entropy']


class _FusedCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target, smoothing):
        ext = require_ext()
        loss, lse = ext.ce_loss_fwd(logits, target, smoothing)
        ctx.save_for_backward(logits, target, lse)
        ctx.smoothing = smoothing
        return loss

    @staticmethod
    def backward(ctx, dloss):
        ext = require_ext()
        logits, target, lse = ctx.saved_tensors
        dlogits = ext.ce_loss_bwd(logits, target, lse, dloss.contiguous(), ctx.smoothing)
        dtarget = None
        if ctx.needs_input_grad[1] and target.dim() == 2:
            # d/dt [lse * sum_t - dot(t, z)] = lse - z
            dtarget = (lse.unsqueeze(1) - logits.float()).to(target.dtype) * dloss.unsqueeze(1)
        return dlogits, dtarget, None


def _cpu_reference(logits: torch.Tensor, target: torch.Tensor, smoothing: float) -> torch.Tensor:
    logprobs = F.log_softmax(logits.float(), dim=-1)
    if target.dim() == 2:
        return torch.sum(-target.float() * logprobs, dim=-1)
    nll = -logprobs.gather(1, target.unsqueeze(1)).squeeze(1)
    if smoothing > 0.:
        smooth = -logprobs.mean(dim=-1)
        return (1. - smoothing) * nll + smoothing * smooth
    return nll


def fused_cross_entropy(
        logits: torch.Tensor,
        target: torch.Tensor,
        smoothing: float = 0.0,
        reduction: str = 'mean',
) -> torch.Tensor:
    """Cross-entropy with optional label smoothing (int64 target) or soft
    targets (float target of logits' shape). Returns per-row loss for
    reduction='none'."""
    if use_hip(logits):
        loss = _FusedCEFn.apply(logits, target, float(smoothing))
    else:
        loss = _cpu_reference(logits, target, smoothing)
    if reduction == 'mean':
        return loss.mean()
    if reduction == 'sum':
        return loss.sum()
    return loss
