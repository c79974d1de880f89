"""Scale-invariance projection shared by AdamP / SGDP (arxiv 2006.08217).

For weights whose function is invariant to scaling (conv filters / linear
rows followed by normalization), the radial component of the update only
changes the effective step size.  AdamP/SGDP detect near-scale-invariant
parameters by cosine similarity and project that radial component out,
also suppressing weight decay for them (wd_ratio).
"""
import math

import torch


def _per_channel(t: torch.Tensor) -> torch.Tensor:
    return t.reshape(t.shape[0], -1)


def _flat(t: torch.Tensor) -> torch.Tensor:
    return t.reshape(1, -1)


def _cosine(a: torch.Tensor, b: torch.Tensor, eps: float) -> torch.Tensor:
    a = a / a.norm(dim=1, keepdim=True).add_(eps)
    b = b / b.norm(dim=1, keepdim=True).add_(eps)
    return (a * b).sum(dim=1).abs_()


def project_scale_invariant(p, grad, perturb, delta: float, wd_ratio: float, eps: float):
    """Remove the radial component of ``perturb`` when p looks scale-invariant.

    Returns (projected perturb, effective weight-decay ratio).
    """
    wd = 1.
    expand_shape = (-1,) + (1,) * (p.dim() - 1)
    for view in (_per_channel, _flat):
        cos = _cosine(view(grad), view(p.data), eps)
        if cos.max() < delta / math.sqrt(view(p.data).shape[1]):
            p_n = p.data / view(p.data).norm(dim=1).add_(eps).reshape(expand_shape)
            radial = (view(p_n) * view(perturb)).sum(dim=1).reshape(expand_shape)
            perturb = perturb - p_n * radial
            return perturb, wd_ratio
    return perturb, wd
