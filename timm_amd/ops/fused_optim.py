"""Multi-tensor fused optimizer primitives.

Replaces the reference's `torch._foreach_*` optimizer hot loops
(`timm/optim/adamw.py:180`, `timm/utils/model_ema.py:227-231`) with single
HIP multi-tensor kernels: one launch updates every parameter chunk of the
model (86M-305M elements for the benchmark models).

CPU / no-ext path: torch._foreach_* composition (same math).
"""
import math
from typing import List, Optional

import torch

from . import _load_extension

_CHUNK = 1 << 20  # elements per kernel chunk entry


def fused_adamw_step(
        params: List[torch.Tensor],
        grads: List[torch.Tensor],
        exp_avgs: List[torch.Tensor],
        exp_avg_sqs: List[torch.Tensor],
        lr: float,
        beta1: float,
        beta2: float,
        eps: float,
        weight_decay: float,
        step: int,
        caution: bool = False,
) -> None:
    """AdamW step over a flat list of same-device tensors (in-place)."""
    if not params:
        return
    bias_correction1 = 1 - beta1 ** step
    bias_correction2 = 1 - beta2 ** step
    ext = _load_extension()
    if params[0].is_cuda and ext is not None and not caution:
        ext.multi_tensor_adamw(
            params, grads, exp_avgs, exp_avg_sqs,
            lr, beta1, beta2, eps, weight_decay, bias_correction1, bias_correction2)
        return
    # foreach reference path
    if weight_decay != 0:
        torch._foreach_mul_(params, 1 - lr * weight_decay)
    torch._foreach_lerp_(exp_avgs, grads, 1 - beta1)
    torch._foreach_mul_(exp_avg_sqs, beta2)
    torch._foreach_addcmul_(exp_avg_sqs, grads, grads, 1 - beta2)
    step_size = lr / bias_correction1
    denom = torch._foreach_sqrt(exp_avg_sqs)
    torch._foreach_div_(denom, math.sqrt(bias_correction2))
    torch._foreach_add_(denom, eps)
    if caution:
        # "cautious" variant: zero the update where sign(update) != sign(grad)
        upd = torch._foreach_div(exp_avgs, denom)
        for p, u, g in zip(params, upd, grads):
            mask = (u * g > 0).to(g.dtype)
            mask.div_(mask.mean().clamp_(min=1e-3))
            p.add_(u * mask, alpha=-step_size)
    else:
        torch._foreach_addcdiv_(params, exp_avgs, denom, -step_size)


def fused_lerp_(
        dsts: List[torch.Tensor],
        srcs: List[torch.Tensor],
        weight: float,
) -> None:
    """dst <- dst + weight * (src - dst); the EMA update (`model_ema.py:227-231`)."""
    if not dsts:
        return
    ext = _load_extension()
    if dsts[0].is_cuda and ext is not None:
        ext.multi_tensor_lerp(dsts, srcs, weight)
        return
    torch._foreach_lerp_(dsts, srcs, weight)


def fused_l2norm(tensors: List[torch.Tensor]) -> torch.Tensor:
    """Global L2 norm over a tensor list (grad clipping, `utils/clip_grad.py:6`)."""
    if not tensors:
        return torch.tensor(0.)
    ext = _load_extension()
    if tensors[0].is_cuda and ext is not None:
        return ext.multi_tensor_l2norm(tensors)
    norms = torch._foreach_norm(tensors)
    return torch.linalg.vector_norm(torch.stack([n.float() for n in norms]))


def fused_lamb_step(
        params: List[torch.Tensor],
        grads: List[torch.Tensor],
        exp_avgs: List[torch.Tensor],
        exp_avg_sqs: List[torch.Tensor],
        lr: float,
        beta1: float,
        beta2: float,
        beta3: float,
        eps: float,
        weight_decay: float,
        bias_correction1: float,
        bias_correction2: float,
        clip_norm: Optional[float] = None,
        always_adapt: bool = False,
        trust_clip: bool = False,
) -> bool:
    """LAMB step over a flat list of same-device tensors: two kernel
    launches for the whole group (m/v update + per-tensor trust-ratio
    scaling; `csrc/multi_tensor.hip`). Returns False when the fused path is
    unavailable and the caller must run the composable step."""
    if not params:
        return True
    ext = _load_extension()
    if not params[0].is_cuda or ext is None:
        return False
    adapt = weight_decay != 0 or always_adapt
    ext.multi_tensor_lamb(
        params, grads, exp_avgs, exp_avg_sqs,
        lr, beta1, beta2, beta3, eps, weight_decay,
        bias_correction1, bias_correction2,
        clip_norm if clip_norm is not None else 0.0,
        adapt, trust_clip)
    return True
