"""NAdamW optimizer (reference `timm/optim/nadamw.py`) — AdamW w/ Nesterov momentum."""
import math
from typing import List, Optional, Tuple

import torch
from torch import Tensor


class NAdamW(torch.optim.Optimizer):
    """Based on the NAdamW of the MLCommons algorithmic-efficiency baselines."""

    def __init__(
            self,
            params,
            lr: float = 1e-3,
            betas: Tuple[float, float] = (0.9, 0.999),
            eps: float = 1e-8,
            weight_decay: float = 1e-2,
            caution: bool = False,
            maximize: bool = False,
    ):
        if not 0.0 <= lr:
            raise ValueError(f'Invalid learning rate: {lr}')
        if not 0.0 <= eps:
            raise ValueError(f'Invalid epsilon value: {eps}')
        if not 0.0 <= betas[0] < 1.0:
            raise ValueError(f'Invalid beta parameter at index 0: {betas[0]}')
        if not 0.0 <= betas[1] < 1.0:
            raise ValueError(f'Invalid beta parameter at index 1: {betas[1]}')
        if not 0.0 <= weight_decay:
            raise ValueError(f'Invalid weight_decay value: {weight_decay}')
        defaults = dict(
            lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
            caution=caution, maximize=maximize,
        )
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('caution', False)
            group.setdefault('maximize', False)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            beta1, beta2 = group['betas']
            for p in group['params']:
                if p.grad is None:
                    continue
                grad = p.grad if not group['maximize'] else -p.grad
                state = self.state[p]
                if len(state) == 0:
                    state['step'] = 0
                    state['exp_avg'] = torch.zeros_like(p)
                    state['exp_avg_sq'] = torch.zeros_like(p)
                state['step'] += 1
                step = state['step']
                exp_avg, exp_avg_sq = state['exp_avg'], state['exp_avg_sq']

                # decoupled decay
                p.mul_(1. - group['lr'] * group['weight_decay'])

                exp_avg.lerp_(grad, 1. - beta1)
                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1. - beta2)

                bias_correction1 = 1 - beta1 ** step
                bias_correction2 = 1 - beta2 ** step
                step_size = group['lr'] / bias_correction1
                bias_correction2_sqrt = math.sqrt(bias_correction2)

                # Nesterov-style momentum: interpolate grad into the first moment
                nesterov_m = exp_avg.mul(beta1).add_(grad, alpha=1. - beta1)

                denom = (exp_avg_sq.sqrt() / bias_correction2_sqrt).add_(group['eps'])

                if group['caution']:
                    update = nesterov_m / denom
                    mask = (update * grad > 0).to(grad.dtype)
                    mask.div_(mask.mean().clamp_(min=1e-3))
                    p.add_(update * mask, alpha=-step_size)
                else:
                    p.addcdiv_(nesterov_m, denom, value=-step_size)

        return loss
