"""NovoGrad (arxiv 1905.11286): layer-wise second moments.

Behavioral parity: /root/reference/timm/optim/nvnovograd.py.  The second
moment is a SCALAR per tensor (||g||^2 EMA); weight decay is folded into the
momentum term after normalization.
"""
import torch
from torch.optim.optimizer import Optimizer

__all__ = ['NvNovoGrad']


class NvNovoGrad(Optimizer):
    def __init__(
            self,
            params,
            lr=1e-3,
            betas=(0.95, 0.98),
            eps=1e-8,
            weight_decay=0,
            grad_averaging=False,
            amsgrad=False,
    ):
        defaults = dict(
            lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
            grad_averaging=grad_averaging, amsgrad=amsgrad)
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('amsgrad', False)

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
                if grad.is_sparse:
                    raise RuntimeError('Sparse gradients are not supported.')
                state = self.state[p]
                if len(state) == 0:
                    state['step'] = 0
                    state['exp_avg'] = torch.zeros_like(p)
                    state['exp_avg_sq'] = torch.zeros([]).to(p.device)
                    if group['amsgrad']:
                        state['max_exp_avg_sq'] = torch.zeros([]).to(p.device)

                exp_avg, exp_avg_sq = state['exp_avg'], state['exp_avg_sq']
                state['step'] += 1

                norm_sq = torch.sum(grad * grad)
                if exp_avg_sq == 0:
                    exp_avg_sq.copy_(norm_sq)
                else:
                    exp_avg_sq.mul_(beta2).add_(norm_sq, alpha=1 - beta2)

                if group['amsgrad']:
                    max_v = state['max_exp_avg_sq']
                    torch.max(max_v, exp_avg_sq, out=max_v)
                    denom = max_v.sqrt().add_(group['eps'])
                else:
                    denom = exp_avg_sq.sqrt().add_(group['eps'])

                grad = grad / denom
                if group['weight_decay'] != 0:
                    grad = grad.add(p, alpha=group['weight_decay'])
                if group['grad_averaging']:
                    grad = grad * (1 - beta1)
                exp_avg.mul_(beta1).add_(grad)

                p.add_(exp_avg, alpha=-group['lr'])

        return loss
