"""RAdam — Adam with rectified adaptive learning rate warmup (reference
`timm/optim/radam.py`; paper arxiv 1908.03265).

While the variance estimate is still unreliable (SMA length < 5) the update
falls back to plain momentum SGD; once it stabilizes, a rectification factor
scales the adaptive step. Per-step coefficients are cached in a 10-slot ring
buffer shared by all params at the same step count.
"""
import math

import torch
from torch.optim.optimizer import Optimizer


class RAdam(Optimizer):
    def __init__(
            self,
            params,
            lr=1e-3,
            betas=(0.9, 0.999),
            eps=1e-8,
            weight_decay=0,
            caution=False,
    ):
        super().__init__(params, dict(
            lr=lr, betas=betas, eps=eps, weight_decay=weight_decay, caution=caution,
            buffer=[[None, None, None] for _ in range(10)]))

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('caution', False)

    def _step_coeffs(self, group, step):
        """(num_sma, step_size) for this step, memoized in the ring buffer."""
        slot = group['buffer'][step % 10]
        if slot[0] == step:
            return slot[1], slot[2]
        beta1, beta2 = group['betas']
        beta2_t = beta2 ** step
        sma_max = 2 / (1 - beta2) - 1
        num_sma = sma_max - 2 * step * beta2_t / (1 - beta2_t)
        if num_sma >= 5:
            rect = math.sqrt(
                (1 - beta2_t)
                * (num_sma - 4) / (sma_max - 4)
                * (num_sma - 2) / num_sma
                * sma_max / (sma_max - 2))
            step_size = group['lr'] * rect / (1 - beta1 ** step)
        else:
            step_size = group['lr'] / (1 - beta1 ** step)
        slot[0], slot[1], slot[2] = step, num_sma, step_size
        return num_sma, step_size

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
                grad = p.grad.float()
                if grad.is_sparse:
                    raise RuntimeError('RAdam does not support sparse gradients')
                p_fp32 = p.float()

                state = self.state[p]
                if len(state) == 0:
                    state['step'] = 0
                    state['exp_avg'] = torch.zeros_like(p_fp32)
                    state['exp_avg_sq'] = torch.zeros_like(p_fp32)
                else:
                    state['exp_avg'] = state['exp_avg'].type_as(p_fp32)
                    state['exp_avg_sq'] = state['exp_avg_sq'].type_as(p_fp32)
                exp_avg, exp_avg_sq = state['exp_avg'], state['exp_avg_sq']

                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
                state['step'] += 1

                num_sma, step_size = self._step_coeffs(group, state['step'])

                if group['weight_decay'] != 0:
                    p_fp32.add_(p_fp32, alpha=-group['weight_decay'] * group['lr'])

                if num_sma >= 5:
                    denom = exp_avg_sq.sqrt().add_(group['eps'])
                    if group['caution']:
                        # zero components whose sign disagrees with the raw grad
                        update = exp_avg / denom
                        mask = (update * grad > 0).to(grad.dtype)
                        mask.div_(mask.mean().clamp_(min=1e-3))
                        p_fp32.add_(update * mask, alpha=-step_size)
                    else:
                        p_fp32.addcdiv_(exp_avg, denom, value=-step_size)
                else:
                    p_fp32.add_(exp_avg, alpha=-step_size)

                p.copy_(p_fp32)

        return loss
