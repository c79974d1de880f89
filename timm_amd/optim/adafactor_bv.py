"""Adafactor, Big-Vision flavour (scaling-ViT / navit training recipes).

Behavioral parity: /root/reference/timm/optim/adafactor_bv.py:49 — the
big_vision variant of Adafactor: factored second moments for large dims,
step-scheduled beta2 (t^-decay_rate_pow), RMS update clipping, optional
low-precision momentum.
"""
import math
from typing import Optional, Tuple

import torch
from torch.optim import Optimizer

__all__ = ['AdafactorBigVision']


def _factored_dims(shape, factored: bool, min_dim_size_to_factor: int) -> Optional[Tuple[int, int]]:
    """Pick the two largest dims to factor over, or None when not factoring."""
    if not factored or len(shape) < 2:
        return None
    sorted_dims = sorted(((d, i) for i, d in enumerate(shape)))
    if sorted_dims[-2][0] < min_dim_size_to_factor:
        return None
    return int(sorted_dims[-2][1]), int(sorted_dims[-1][1])


class AdafactorBigVision(Optimizer):
    def __init__(
            self,
            params,
            lr: float = 1.0,
            min_dim_size_to_factor: int = 16,
            decay_rate: float = 0.8,
            decay_offset: int = 0,
            beta2_cap: float = 0.999,
            momentum: Optional[float] = 0.9,
            momentum_dtype: torch.dtype = torch.bfloat16,
            eps: Optional[float] = None,
            weight_decay: float = 0.0,
            clipping_threshold: Optional[float] = 1.0,
            unscaled_wd: bool = False,
            caution: bool = False,
            *,
            foreach: Optional[bool] = False,
    ):
        defaults = dict(
            lr=lr,
            min_dim_size_to_factor=min_dim_size_to_factor,
            decay_rate=decay_rate,
            decay_offset=decay_offset,
            beta2_cap=beta2_cap,
            momentum=momentum,
            momentum_dtype=momentum_dtype,
            eps=eps,
            weight_decay=weight_decay,
            clipping_threshold=clipping_threshold,
            unscaled_wd=unscaled_wd,
            caution=caution,
            foreach=foreach,
        )
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('caution', False)

    @staticmethod
    def _beta2(step: int, decay_rate: float, cap: float) -> float:
        """Step-scheduled beta2: 1 - t^-decay_rate, capped."""
        return min(cap, 1.0 - step ** -decay_rate)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            for p in group['params']:
                if p.grad is None:
                    continue
                grad = p.grad
                state = self.state[p]

                if len(state) == 0:
                    state['step'] = 0
                    factored = _factored_dims(
                        p.shape, factored=True,
                        min_dim_size_to_factor=group['min_dim_size_to_factor'])
                    if factored is not None:
                        dc, dr = factored
                        row_shape = list(p.shape)
                        row_shape[dr] = 1
                        col_shape = list(p.shape)
                        col_shape[dc] = 1
                        state['exp_avg_sq_r'] = p.new_zeros(row_shape)
                        state['exp_avg_sq_c'] = p.new_zeros(col_shape)
                        state['factored'] = (dc, dr)
                    else:
                        state['exp_avg_sq'] = torch.zeros_like(p, memory_format=torch.preserve_format)
                        state['factored'] = None
                    if group['momentum'] is not None:
                        state['exp_avg'] = torch.zeros_like(p, dtype=group['momentum_dtype'])

                state['step'] += 1
                step = state['step'] - group['decay_offset']
                beta2 = self._beta2(max(step, 1), group['decay_rate'], group['beta2_cap'])
                one_minus = 1 - beta2

                grad_sq = grad * grad
                if group['eps'] is not None:
                    grad_sq = grad_sq + group['eps']

                if state['factored'] is not None:
                    dc, dr = state['factored']
                    vr = state['exp_avg_sq_r']
                    vc = state['exp_avg_sq_c']
                    vr.mul_(beta2).add_(grad_sq.mean(dim=dr, keepdim=True), alpha=one_minus)
                    vc.mul_(beta2).add_(grad_sq.mean(dim=dc, keepdim=True), alpha=one_minus)
                    # rank-1 reconstruction, normalized by the shared mean
                    reduce_dc = dc - 1 if dc > dr else dc
                    row_col_mean = vr.mean(dim=reduce_dc, keepdim=True)
                    row_factor = (vr / row_col_mean.clamp(min=1e-30)).rsqrt()
                    col_factor = vc.rsqrt()
                    update = grad * row_factor * col_factor
                else:
                    v = state['exp_avg_sq']
                    v.mul_(beta2).add_(grad_sq, alpha=one_minus)
                    update = grad * v.rsqrt()

                if group['clipping_threshold'] is not None:
                    # clip the update RMS at the threshold
                    rms = update.norm(2) / math.sqrt(update.numel())
                    update = update / (rms / group['clipping_threshold']).clamp(min=1.0)

                if group['momentum'] is not None:
                    m = state['exp_avg']
                    m.mul_(group['momentum']).add_(
                        update.to(group['momentum_dtype']), alpha=1 - group['momentum'])
                    update = m.to(p.dtype)
                    if group['caution']:
                        mask = (update * grad > 0).to(grad.dtype)
                        mask.div_(mask.mean().clamp_(min=1e-3))
                        update = update * mask

                wd = group['weight_decay']
                if wd != 0:
                    if group['unscaled_wd']:
                        p.mul_(1.0 - wd)  # pre-lr-scaled decay (big-vision default)
                    else:
                        p.mul_(1.0 - group['lr'] * wd)

                p.add_(update, alpha=-group['lr'])

        return loss
