"""LAMB optimizer (reference `timm/optim/lamb.py:67`).

Adam moments + per-layer trust ratio ||w||/||u||.  Elementwise work goes
through foreach tensor ops; the trust-ratio norms are per-tensor.
"""
import math
from typing import Optional, Tuple

import torch
from torch.optim import Optimizer


class Lamb(Optimizer):
    """Implements the LAMB algorithm (https://arxiv.org/abs/1904.00962)."""

    def __init__(
            self,
            params,
            lr: float = 1e-3,
            bias_correction: bool = True,
            betas: Tuple[float, float] = (0.9, 0.999),
            eps: float = 1e-6,
            weight_decay: float = 0.01,
            grad_averaging: bool = True,
            max_grad_norm: Optional[float] = 1.0,
            trust_clip: bool = False,
            always_adapt: bool = False,
            caution: bool = False,
            decoupled_decay: bool = False,
    ):
        defaults = dict(
            lr=lr, bias_correction=bias_correction, betas=betas, eps=eps,
            weight_decay=weight_decay, grad_averaging=grad_averaging,
            max_grad_norm=max_grad_norm, trust_clip=trust_clip, always_adapt=always_adapt,
            caution=caution, decoupled_decay=decoupled_decay,
        )
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('caution', False)
            group.setdefault('decoupled_decay', False)

    def _get_clip_grad_norm(self):
        max_grad_norm = self.defaults['max_grad_norm']
        if max_grad_norm is None:
            return None

        norms = []
        for group in self.param_groups:
            for p in group['params']:
                grad = p.grad
                if grad is None:
                    continue
                if grad.is_sparse:
                    raise RuntimeError('Lamb does not support sparse gradients.')
                norms.append(torch.linalg.vector_norm(grad))
        global_norm = torch.linalg.vector_norm(torch.stack(norms))
        clip_global_norm = (global_norm / max_grad_norm).clamp_(min=1.0)
        return clip_global_norm

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        clip_grad_norm = self._get_clip_grad_norm()  # None if disabled

        for group in self.param_groups:
            bias_correction = 1 if group['bias_correction'] else 0
            beta1, beta2 = group['betas']
            grad_averaging = 1 if group['grad_averaging'] else 0
            beta3 = 1 - beta1 if grad_averaging else 1.0

            # assume same step across group now to simplify things
            if 'step' in group:
                group['step'] += 1
            else:
                group['step'] = 1

            if bias_correction:
                bias_correction1 = 1 - beta1 ** group['step']
                bias_correction2 = 1 - beta2 ** group['step']
            else:
                bias_correction1, bias_correction2 = 1.0, 1.0

            # fused multi-tensor path: whole group in two HIP launches
            # (csrc/multi_tensor.hip); the composable loop below stays the
            # reference for CPU and for caution/decoupled-decay variants
            if not group['caution'] and not group.get('decoupled_decay', False):
                group_params, group_grads, group_m, group_v = [], [], [], []
                for p in group['params']:
                    if p.grad is None:
                        continue
                    state = self.state[p]
                    if len(state) == 0:
                        state['exp_avg'] = torch.zeros_like(p, dtype=torch.float32)
                        state['exp_avg_sq'] = torch.zeros_like(p, dtype=torch.float32)
                    if state['exp_avg'].dtype != torch.float32:
                        group_params = None  # pre-existing non-fp32 state: composable path
                        break
                    group_params.append(p)
                    group_grads.append(p.grad)
                    group_m.append(state['exp_avg'])
                    group_v.append(state['exp_avg_sq'])
                from .. import ops
                clip = None
                if clip_grad_norm is not None:
                    clip = float(clip_grad_norm)
                if group_params is not None and group_params and ops.fused_lamb_step(
                        group_params, group_grads, group_m, group_v,
                        group['lr'], beta1, beta2, beta3, group['eps'],
                        group['weight_decay'], bias_correction1, bias_correction2,
                        clip_norm=clip, always_adapt=group['always_adapt'],
                        trust_clip=group['trust_clip']):
                    continue

            for p in group['params']:
                if p.grad is None:
                    continue
                grad = p.grad

                if clip_grad_norm is not None:
                    grad.div_(clip_grad_norm)

                state = self.state[p]
                if len(state) == 0:
                    state['exp_avg'] = torch.zeros_like(p)
                    state['exp_avg_sq'] = torch.zeros_like(p)

                exp_avg, exp_avg_sq = state['exp_avg'], state['exp_avg_sq']

                exp_avg.mul_(beta1).add_(grad, alpha=beta3)  # m_t
                exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)  # v_t

                denom = (exp_avg_sq.sqrt() / math.sqrt(bias_correction2)).add_(group['eps'])
                update = (exp_avg / bias_correction1).div_(denom)

                if group['caution']:
                    mask = (update * grad > 0).to(grad.dtype)
                    mask.div_(mask.mean().clamp_(min=1e-3))
                    update.mul_(mask)

                weight_decay = group['weight_decay']
                if weight_decay != 0:
                    if group.get('decoupled_decay', False):
                        p.add_(p, alpha=-group['lr'] * weight_decay)
                    else:
                        update.add_(p, alpha=weight_decay)

                if weight_decay != 0 or group['always_adapt']:
                    # Layer-wise LR adaptation. Skip adaptation on parameters that are
                    # excluded from weight decay, unless always_adapt == True
                    w_norm = p.norm(2.0)
                    g_norm = update.norm(2.0)
                    trust_ratio = w_norm / g_norm
                    # FIXME nested where required since logical and/or not working in PT XLA
                    # Set the ratio to 1.0 (no change) if either weight norm or grad norm is zero
                    trust_ratio = torch.where(
                        w_norm > 0,
                        torch.where(g_norm > 0, trust_ratio, 1.0),
                        1.0,
                    )
                    if group['trust_clip']:
                        trust_ratio = torch.clamp(trust_ratio, max=1.0)
                    update.mul_(trust_ratio)

                p.add_(update, alpha=-group['lr'])

        return loss
