"""Adaptive gradient clipping — clip each unit's gradient relative to its
parameter norm (NFNets, arxiv 2102.06171; reference `timm/utils/agc.py:30`)."""
import torch


def unitwise_norm(x, norm_type=2.0):
    """Per-output-unit norm: scalar for vectors/biases, per-row (first dim)
    for Conv/Linear kernels."""
    if x.ndim <= 1:
        return x.norm(norm_type)
    return x.norm(norm_type, dim=tuple(range(1, x.ndim)), keepdim=True)


def adaptive_clip_grad(parameters, clip_factor=0.01, eps=1e-3, norm_type=2.0):
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    for p in parameters:
        if p.grad is None:
            continue
        w = p.detach()
        g = p.grad.detach()
        limit = unitwise_norm(w, norm_type=norm_type).clamp_(min=eps).mul_(clip_factor)
        g_norm = unitwise_norm(g, norm_type=norm_type)
        scaled = g * (limit / g_norm.clamp(min=1e-6))
        p.grad.detach().copy_(torch.where(g_norm < limit, g, scaled))
