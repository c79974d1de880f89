"""AdamP (arxiv 2006.08217): Adam with the scale-invariance projection.

Behavioral parity: /root/reference/timm/optim/adamp.py.  The projection lives
in the shared `_projection` module (also used by SGDP).
"""
import math

import torch
from torch.optim.optimizer import Optimizer

from ._projection import project_scale_invariant

__all__ = ['AdamP']


class AdamP(Optimizer):
    def __init__(
            self,
            params,
            lr=1e-3,
            betas=(0.9, 0.999),
            eps=1e-8,
            weight_decay=0,
            delta=0.1,
            wd_ratio=0.1,
            nesterov=False,
    ):
        defaults = dict(
            lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
            delta=delta, wd_ratio=wd_ratio, nesterov=nesterov)
        super().__init__(params, defaults)

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

                state['step'] += 1
                bc1 = 1 - beta1 ** state['step']
                bc2 = 1 - beta2 ** state['step']

                exp_avg, exp_avg_sq = state['exp_avg'], state['exp_avg_sq']
                exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)

                denom = (exp_avg_sq.sqrt() / math.sqrt(bc2)).add_(group['eps'])
                step_size = group['lr'] / bc1
                if group['nesterov']:
                    perturb = (beta1 * exp_avg + (1 - beta1) * grad) / denom
                else:
                    perturb = exp_avg / denom

                wd_ratio = 1.
                if len(p.shape) > 1:
                    perturb, wd_ratio = project_scale_invariant(
                        p, grad, perturb, group['delta'], group['wd_ratio'], group['eps'])

                if group['weight_decay'] > 0:
                    p.mul_(1 - group['lr'] * group['weight_decay'] * wd_ratio)
                p.add_(perturb, alpha=-step_size)

        return loss
