"""MADGRAD optimizer (reference `timm/optim/madgrad.py`;
paper https://arxiv.org/abs/2101.11075)."""
import math
from typing import TYPE_CHECKING, Any, Callable, Optional

import torch
import torch.optim


class MADGRAD(torch.optim.Optimizer):
    """A Momentumized, Adaptive, Dual Averaged Gradient Method for Stochastic Optimization."""

    def __init__(
            self,
            params,
            lr: float = 1e-2,
            momentum: float = 0.9,
            weight_decay: float = 0,
            eps: float = 1e-6,
            decoupled_decay: bool = False,
    ):
        if momentum < 0 or momentum >= 1:
            raise ValueError(f"Momentum {momentum} must be in the range [0,1]")
        if lr <= 0:
            raise ValueError(f"Learning rate {lr} must be positive")
        if weight_decay < 0:
            raise ValueError(f"Weight decay {weight_decay} must be non-negative")
        if eps < 0:
            raise ValueError(f"Eps must be non-negative")

        defaults = dict(
            lr=lr, eps=eps, momentum=momentum, weight_decay=weight_decay, decoupled_decay=decoupled_decay)
        super().__init__(params, defaults)

    @property
    def supports_memory_efficient_fp16(self) -> bool:
        return False

    @property
    def supports_flat_params(self) -> bool:
        return True

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            eps = group['eps']
            lr = group['lr'] + eps
            weight_decay = group['weight_decay']
            momentum = group['momentum']
            ck = 1 - momentum

            for p in group["params"]:
                if p.grad is None:
                    continue
                grad = p.grad
                if momentum != 0.0 and grad.is_sparse:
                    raise RuntimeError("momentum != 0 is not compatible with sparse gradients")

                state = self.state[p]
                if len(state) == 0:
                    state['step'] = 0
                    state['grad_sum_sq'] = torch.zeros_like(p)
                    state['s'] = torch.zeros_like(p)
                    if momentum != 0:
                        state['x0'] = torch.clone(p).detach()

                state['step'] += 1
                grad_sum_sq = state['grad_sum_sq']
                s = state['s']
                lamb = lr * math.sqrt(state['step'])

                # Apply weight decay
                if weight_decay != 0:
                    if group['decoupled_decay']:
                        p.mul_(1.0 - group['lr'] * weight_decay)
                    else:
                        if grad.is_sparse:
                            raise RuntimeError("weight_decay option is not compatible with sparse gradients")
                        grad.add_(p, alpha=weight_decay)

                if momentum == 0:
                    # Compute x_0 from other known quantities
                    rms = grad_sum_sq.pow(1 / 3).add_(eps)
                    x0 = p.addcdiv(s, rms, value=1)
                else:
                    x0 = state['x0']

                # Accumulate second moments
                grad_sum_sq.addcmul_(grad, grad, value=lamb)
                rms = grad_sum_sq.pow(1 / 3).add_(eps)

                # Update s
                s.add_(grad, alpha=lamb)

                # Step
                if momentum == 0:
                    p.copy_(x0.addcdiv(s, rms, value=-1))
                else:
                    z = x0.addcdiv(s, rms, value=-1)
                    # p is a moving average of z
                    p.mul_(1 - ck).add_(z, alpha=ck)

        return loss
