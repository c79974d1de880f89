"""LARS / LARC (reference `timm/optim/lars.py:17`; papers arxiv 1708.03888 /
LARC from NVIDIA apex).

SGD with a per-tensor trust ratio: the effective step is scaled by
`trust_coeff * ||w|| / (||g|| + wd * ||w||)`, optionally clipped at 1 relative
to the base LR (LARC). Large-batch training keeps layer updates proportional
to layer magnitude.
"""
import torch
from torch.optim.optimizer import Optimizer


class Lars(Optimizer):
    def __init__(
            self,
            params,
            lr=1.0,
            momentum=0,
            dampening=0,
            weight_decay=0,
            nesterov=False,
            trust_coeff=0.001,
            eps=1e-8,
            trust_clip=False,
            always_adapt=False,
    ):
        if lr < 0.0:
            raise ValueError(f'Invalid learning rate: {lr}')
        if momentum < 0.0:
            raise ValueError(f'Invalid momentum value: {momentum}')
        if weight_decay < 0.0:
            raise ValueError(f'Invalid weight_decay value: {weight_decay}')
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError('Nesterov momentum requires a momentum and zero dampening')
        super().__init__(params, dict(
            lr=lr,
            momentum=momentum,
            dampening=dampening,
            weight_decay=weight_decay,
            nesterov=nesterov,
            trust_coeff=trust_coeff,
            eps=eps,
            trust_clip=trust_clip,
            always_adapt=always_adapt,
        ))

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('nesterov', False)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            wd = group['weight_decay']
            momentum = group['momentum']
            dampening = group['dampening']

            for p in group['params']:
                if p.grad is None:
                    continue
                grad = p.grad

                if wd != 0 or group['always_adapt']:
                    w_norm = p.norm(2.0)
                    g_norm = grad.norm(2.0)
                    ratio = group['trust_coeff'] * w_norm / (g_norm + w_norm * wd + group['eps'])
                    # nested where: either norm being zero disables adaptation
                    ratio = torch.where(
                        w_norm > 0,
                        torch.where(g_norm > 0, ratio, 1.0),
                        1.0,
                    )
                    if group['trust_clip']:
                        # LARC: cap the adapted LR at the base LR
                        ratio = torch.clamp(ratio / group['lr'], max=1.0)
                    grad.add_(p, alpha=wd)
                    grad.mul_(ratio)

                if momentum != 0:
                    state = self.state[p]
                    buf = state.get('momentum_buffer')
                    if buf is None:
                        state['momentum_buffer'] = buf = torch.clone(grad).detach()
                    else:
                        buf.mul_(momentum).add_(grad, alpha=1. - dampening)
                    grad = grad.add(buf, alpha=momentum) if group['nesterov'] else buf

                p.add_(grad, alpha=-group['lr'])

        return loss
