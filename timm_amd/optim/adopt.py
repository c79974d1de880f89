"""ADOPT optimizer (reference `timm/optim/adopt.py:59`;
paper: ADOPT: Modified Adam Can Converge with Any β2, https://arxiv.org/abs/2411.02853)."""
from typing import Tuple

import torch
from torch.optim.optimizer import Optimizer


class Adopt(Optimizer):
    def __init__(
            self,
            params,
            lr: float = 1e-3,
            betas: Tuple[float, float] = (0.9, 0.9999),
            eps: float = 1e-6,
            clip_exp: float = 0.25,
            weight_decay: float = 0.0,
            decoupled: bool = False,
            caution: bool = False,
    ):
        defaults = dict(
            lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
            clip_exp=clip_exp, decoupled=decoupled, caution=caution,
        )
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('caution', False)
            group.setdefault('clip_exp', None)

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
                grad = p.grad
                state = self.state[p]
                if len(state) == 0:
                    state['step'] = 0
                    state['exp_avg'] = torch.zeros_like(p)
                    state['exp_avg_sq'] = torch.zeros_like(p)

                exp_avg, exp_avg_sq = state['exp_avg'], state['exp_avg_sq']
                step = state['step']

                if group['weight_decay'] != 0:
                    if group['decoupled']:
                        p.mul_(1. - group['lr'] * group['weight_decay'])
                    else:
                        grad = grad.add(p, alpha=group['weight_decay'])

                if step == 0:
                    # first step initializes v_0 = g_0^2, no param update
                    exp_avg_sq.addcmul_(grad, grad)
                    state['step'] += 1
                    continue

                denom = torch.clamp(exp_avg_sq.sqrt(), group['eps'])
                normed_grad = grad.div(denom)
                if group['clip_exp'] is not None:
                    clip_val = (step - 1) ** group['clip_exp']
                    normed_grad.clamp_(-clip_val, clip_val)

                exp_avg.lerp_(normed_grad, 1. - beta1)

                if group['caution']:
                    mask = (exp_avg * grad > 0).to(grad.dtype)
                    mask.div_(mask.mean().clamp_(min=1e-3))
                    p.add_(exp_avg * mask, alpha=-group['lr'])
                else:
                    p.add_(exp_avg, alpha=-group['lr'])

                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1. - beta2)
                state['step'] += 1

        return loss
