"""SGDP (arxiv 2006.08217): SGD-momentum with the scale-invariance projection.

Behavioral parity: /root/reference/timm/optim/sgdp.py; the projection is in
the shared `_projection` module (also used by AdamP).
"""
import torch
from torch.optim.optimizer import Optimizer

from ._projection import project_scale_invariant

__all__ = ['SGDP']


class SGDP(Optimizer):
    def __init__(
            self,
            params,
            lr=0.1,
            momentum=0,
            dampening=0,
            weight_decay=0,
            nesterov=False,
            eps=1e-8,
            delta=0.1,
            wd_ratio=0.1,
    ):
        defaults = dict(
            lr=lr, momentum=momentum, dampening=dampening, weight_decay=weight_decay,
            nesterov=nesterov, eps=eps, delta=delta, wd_ratio=wd_ratio)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            momentum = group['momentum']
            for p in group['params']:
                if p.grad is None:
                    continue
                grad = p.grad
                state = self.state[p]
                if len(state) == 0:
                    state['momentum'] = torch.zeros_like(p)

                buf = state['momentum']
                buf.mul_(momentum).add_(grad, alpha=1 - group['dampening'])
                if group['nesterov']:
                    perturb = grad + momentum * buf
                else:
                    perturb = buf

                wd_ratio = 1.
                if len(p.shape) > 1:
                    perturb, wd_ratio = project_scale_invariant(
                        p, grad, perturb, group['delta'], group['wd_ratio'], group['eps'])

                if group['weight_decay'] > 0:
                    p.mul_(
                        1 - group['lr'] * group['weight_decay'] * wd_ratio / (1 - momentum))
                p.add_(perturb, alpha=-group['lr'])

        return loss
