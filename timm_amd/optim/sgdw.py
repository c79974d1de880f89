"""SGD with decoupled weight decay (SGDW), reference `timm/optim/sgdw.py`."""
from typing import List, Optional

import torch
from torch import Tensor
from torch.optim.optimizer import Optimizer


class SGDW(Optimizer):
    def __init__(
            self,
            params,
            lr=1e-3,
            momentum=0,
            dampening=0,
            weight_decay=0,
            nesterov=False,
            caution=False,
            maximize: bool = False,
    ):
        if lr < 0.0:
            raise ValueError(f"Invalid learning rate: {lr}")
        if momentum < 0.0:
            raise ValueError(f"Invalid momentum value: {momentum}")

        defaults = dict(
            lr=lr, momentum=momentum, dampening=dampening,
            weight_decay=weight_decay, nesterov=nesterov,
            caution=caution, maximize=maximize)
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError("Nesterov momentum requires a momentum and zero dampening")
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('nesterov', False)
            group.setdefault('caution', False)
            group.setdefault('maximize', False)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            lr = group['lr']
            momentum = group['momentum']
            dampening = group['dampening']
            nesterov = group['nesterov']
            maximize = group['maximize']

            for p in group['params']:
                if p.grad is None:
                    continue
                grad = p.grad if not maximize else -p.grad

                # decoupled decay applied directly to weights
                if group['weight_decay'] != 0:
                    p.mul_(1. - lr * group['weight_decay'])

                if momentum != 0:
                    state = self.state[p]
                    buf = state.get('momentum_buffer', None)
                    if buf is None:
                        buf = torch.clone(grad).detach()
                        state['momentum_buffer'] = buf
                    else:
                        buf.mul_(momentum).add_(grad, alpha=1 - dampening)
                    if nesterov:
                        d_p = grad.add(buf, alpha=momentum)
                    else:
                        d_p = buf
                else:
                    d_p = grad

                if group['caution']:
                    mask = (d_p * grad > 0).to(grad.dtype)
                    mask.div_(mask.mean().clamp_(min=1e-3))
                    d_p = d_p * mask

                p.add_(d_p, alpha=-lr)

        return loss
