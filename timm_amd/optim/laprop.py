"""LaProp (arxiv 2002.04839): momentum on the adaptivity-normalized gradient.

Behavioral parity: /root/reference/timm/optim/laprop.py.  Unlike Adam, the
second-moment normalization is applied to the raw gradient BEFORE it enters
the momentum accumulator, decoupling the two.
"""
import torch
from torch.optim import Optimizer

__all__ = ['LaProp']


class LaProp(Optimizer):
    def __init__(
            self,
            params,
            lr=4e-4,
            betas=(0.9, 0.999),
            eps=1e-15,
            weight_decay=0.,
            caution: bool = False,
    ):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay, caution=caution)
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('caution', False)

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
                    # running sums of the lr-weighted bias corrections
                    state['exp_avg_lr_1'] = 0.
                    state['exp_avg_lr_2'] = 0.

                exp_avg, exp_avg_sq = state['exp_avg'], state['exp_avg_sq']
                state['step'] += 1
                state['exp_avg_lr_1'] = state['exp_avg_lr_1'] * beta1 + (1 - beta1) * group['lr']
                state['exp_avg_lr_2'] = state['exp_avg_lr_2'] * beta2 + (1 - beta2)

                # normalize the gradient by the second moment FIRST
                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
                denom = (exp_avg_sq / state['exp_avg_lr_2']).sqrt_().add_(group['eps'])
                step_of_this_grad = grad / denom

                # then fold into lr-weighted momentum
                exp_avg.mul_(beta1).add_(step_of_this_grad, alpha=(1 - beta1) * group['lr'])
                # bias correction on the lr-weighted first moment
                bc1 = state['exp_avg_lr_1'] / group['lr'] if group['lr'] != 0. else 1.

                update = exp_avg
                if group['caution']:
                    mask = (update * grad > 0).to(grad.dtype)
                    mask.div_(mask.mean().clamp_(min=1e-3))
                    update = update * mask

                p.add_(update, alpha=-1.0 / bc1)
                if group['weight_decay'] != 0:
                    p.add_(p, alpha=-group['weight_decay'] * group['lr'])

        return loss
