"""Gradient clipping dispatch (reference `timm/utils/clip_grad.py:6`).

The global-norm path uses the fused multi-tensor L2 kernel on device.
"""
import torch

from .. import ops
from .agc import adaptive_clip_grad


def clip_grad_norm_fused(parameters, max_norm: float, eps: float = 1e-6):
    """Global-norm clip using the fused multi-tensor kernel."""
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    grads = [p.grad for p in parameters if p.grad is not None]
    if not grads:
        return torch.tensor(0.)
    total_norm = ops.fused_l2norm(grads)
    clip_coef = max_norm / (total_norm + eps)
    if clip_coef < 1:
        torch._foreach_mul_(grads, clip_coef.to(grads[0].dtype))
    return total_norm


def dispatch_clip_grad(parameters, value: float, mode: str = 'norm', norm_type: float = 2.0):
    """Dispatch to gradient clipping method (norm / value / agc)."""
    if mode == 'norm':
        if norm_type == 2.0:
            clip_grad_norm_fused(parameters, value)
        else:
            torch.nn.utils.clip_grad_norm_(parameters, value, norm_type=norm_type)
    elif mode == 'value':
        torch.nn.utils.clip_grad_value_(parameters, value)
    elif mode == 'agc':
        adaptive_clip_grad(parameters, value, norm_type=norm_type)
    else:
        assert False, f"Unknown clip mode ({mode})."
