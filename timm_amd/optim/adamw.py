"""AdamW with a single fused multi-tensor HIP kernel step
(reference `timm/optim/adamw.py:20,180`).

One kernel launch updates every parameter of the model (param dtype bf16 or
fp32; moment states fp32).  `caution` implements the cautious-optimizer
variant (zero updates whose sign disagrees with the gradient).
"""
import math
from typing import List, Optional, Tuple, Union

import torch
from torch.optim.optimizer import Optimizer

from .. import ops


class AdamW(Optimizer):
    def __init__(
            self,
            params,
            lr: float = 1e-3,
            betas: Tuple[float, float] = (0.9, 0.999),
            eps: float = 1e-8,
            weight_decay: float = 1e-2,
            amsgrad: bool = False,
            caution: bool = False,
            corrected_weight_decay: bool = False,
    ):
        if not 0.0 <= lr:
            raise ValueError(f"Invalid learning rate: {lr}")
        defaults = dict(
            lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
            amsgrad=amsgrad, caution=caution, corrected_weight_decay=corrected_weight_decay,
        )
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('amsgrad', False)
            group.setdefault('caution', False)
            group.setdefault('corrected_weight_decay', False)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            params, grads, exp_avgs, exp_avg_sqs = [], [], [], []
            step = None
            for p in group['params']:
                if p.grad is None:
                    continue
                if p.grad.is_sparse:
                    raise RuntimeError('AdamW does not support sparse gradients')
                state = self.state[p]
                if len(state) == 0:
                    state['step'] = 0
                    state['exp_avg'] = torch.zeros_like(p, dtype=torch.float32)
                    state['exp_avg_sq'] = torch.zeros_like(p, dtype=torch.float32)
                state['step'] += 1
                step = state['step']
                params.append(p)
                grads.append(p.grad)
                exp_avgs.append(state['exp_avg'])
                exp_avg_sqs.append(state['exp_avg_sq'])

            if not params:
                continue
            beta1, beta2 = group['betas']
            wd = group['weight_decay']
            if group['corrected_weight_decay']:
                # lr-corrected decay: wd scaled by lr/defaults-lr invariance
                wd = wd * group['lr'] / self.defaults['lr'] if self.defaults['lr'] else wd
            ops.fused_adamw_step(
                params, grads, exp_avgs, exp_avg_sqs,
                lr=group['lr'], beta1=beta1, beta2=beta2, eps=group['eps'],
                weight_decay=wd, step=step, caution=group['caution'],
            )
        return loss
