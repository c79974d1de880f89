"""PSGD-Kron: preconditioned SGD with Kronecker-factored preconditioners
(Xi-Lin Li's PSGD, arxiv 1512.04202 / 2402.11858).

Behavioral parity target: /root/reference/timm/optim/kron.py:82.  Design
differences (documented): gradients with >2 dims are flattened to 2D
(out_chs x rest, like the Muon conv handling) instead of carrying one
Kronecker factor per tensor dim; factors above ``max_size_triangular``
fall back to diagonal preconditioners.  The whitening-criterion update and
probabilistic refresh schedule follow the PSGD reference algorithm.
"""
import math
import random

from typing import Optional

import torch
from torch.optim import Optimizer

__all__ = ['Kron']


def precond_update_prob_schedule(
        n: float, max_prob: float = 1.0, min_prob: float = 0.03,
        decay: float = 0.001, flat_start: float = 500) -> float:
    """Anneal the preconditioner update probability from max to min."""
    if n < flat_start:
        return max_prob
    return max(min_prob, max_prob * math.exp(-decay * (n - flat_start)))


def _norm_lower_bound(a: torch.Tensor) -> torch.Tensor:
    """Cheap lower bound on the spectral norm of PSD matrix ``a``."""
    max_abs = a.diagonal().abs().max()
    if max_abs > 0:
        a = a / max_abs
        return max_abs * a.norm()
    return max_abs


class Kron(Optimizer):
    """PSGD with (left, right) Kronecker preconditioners per 2D gradient.

    Each matrix-shaped gradient G keeps upper-triangular factors Ql, Qr so
    that P = (Ql^T Ql) kron (Qr^T Qr) whitens the gradient distribution.
    1D params (or oversized dims) use diagonal factors.
    """

    def __init__(
            self,
            params,
            lr: float = 0.001,
            momentum: float = 0.9,
            weight_decay: float = 0.0,
            preconditioner_lr: float = 0.1,
            preconditioner_init_scale: float = 1.0,
            max_size_triangular: int = 8192,
            min_ndim_triangular: int = 2,
            mu_dtype: Optional[torch.dtype] = None,
            precond_dtype: Optional[torch.dtype] = None,
            caution: bool = False,
            flatten_conv: bool = True,
            deterministic: bool = False,
    ):
        defaults = dict(
            lr=lr, momentum=momentum, weight_decay=weight_decay,
            preconditioner_lr=preconditioner_lr,
            preconditioner_init_scale=preconditioner_init_scale,
            max_size_triangular=max_size_triangular,
            min_ndim_triangular=min_ndim_triangular,
            mu_dtype=mu_dtype, precond_dtype=precond_dtype,
            caution=caution, flatten_conv=flatten_conv,
        )
        super().__init__(params, defaults)
        self._step_count = 0
        self._rng = random.Random(5318008 if deterministic else None)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('caution', False)

    # -- preconditioner machinery ------------------------------------------
    def _init_factors(self, g2d: torch.Tensor, group, dtype):
        m, n = g2d.shape
        scale = group['preconditioner_init_scale'] ** 0.5
        tri_ok = lambda d: d <= group['max_size_triangular']  # noqa: E731
        ql = (torch.eye(m, device=g2d.device, dtype=dtype) * scale) if tri_ok(m) \
            else torch.full((m,), scale, device=g2d.device, dtype=dtype)
        qr = (torch.eye(n, device=g2d.device, dtype=dtype) * scale) if tri_ok(n) \
            else torch.full((n,), scale, device=g2d.device, dtype=dtype)
        return ql, qr

    @staticmethod
    def _apply_side(q: torch.Tensor, x: torch.Tensor, left: bool) -> torch.Tensor:
        """q @ x (left) or x @ q.T (right); diagonal q broadcasts."""
        if q.dim() == 1:
            return x * q[:, None] if left else x * q[None, :]
        return q @ x if left else x @ q.T

    @staticmethod
    def _solve_side(q: torch.Tensor, x: torch.Tensor, left: bool) -> torch.Tensor:
        """q^-T x (left) or x q^-1 (right) for triangular/diagonal q."""
        if q.dim() == 1:
            return x / q[:, None] if left else x / q[None, :]
        if left:
            return torch.linalg.solve_triangular(q.T.contiguous(), x, upper=False)
        return torch.linalg.solve_triangular(q, x, upper=True, left=False)

    def _update_precond(self, ql, qr, g2d, precond_lr):
        """One whitening-criterion update of (Ql, Qr) from gradient sample G."""
        v = torch.randn_like(g2d)
        a = self._apply_side(ql, self._apply_side(qr, g2d, left=False), left=True)   # Ql G Qr^T
        b = self._solve_side(ql, self._solve_side(qr, v, left=False), left=True)     # Ql^-T V Qr^-1

        for q, left in ((ql, True), (qr, False)):
            if left:
                term1, term2 = a @ a.T, b @ b.T
            else:
                term1, term2 = a.T @ a, b.T @ b
            if q.dim() == 1:
                t1d, t2d = term1.diagonal(), term2.diagonal()
                denom = (t1d + t2d).max().clamp(min=1e-30)
                q.sub_(precond_lr / denom * (t1d - t2d) * q)
            else:
                grad = torch.triu(term1 - term2)
                denom = _norm_lower_bound(term1 + term2).clamp(min=1e-30)
                q.sub_(precond_lr / denom * grad @ q)

    def _precondition(self, ql, qr, g2d):
        """P G = Ql^T Ql G Qr^T Qr."""
        x = self._apply_side(ql, g2d, left=True)
        if ql.dim() == 2:
            x = ql.T @ x
        else:
            x = x * ql[:, None]
        x = self._apply_side(qr, x, left=False)
        if qr.dim() == 2:
            x = x @ qr
        else:
            x = x * qr[None, :]
        return x

    # -- step ---------------------------------------------------------------
    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        self._step_count += 1
        update_prob = precond_update_prob_schedule(self._step_count)
        do_update = self._rng.random() < update_prob

        for group in self.param_groups:
            momentum = group['momentum']
            precond_dtype = group['precond_dtype'] or torch.float32
            for p in group['params']:
                if p.grad is None:
                    continue
                grad = p.grad
                state = self.state[p]

                if len(state) == 0:
                    state['step'] = 0
                    mu_dtype = group['mu_dtype'] or p.dtype
                    state['momentum_buffer'] = torch.zeros_like(p, dtype=mu_dtype)

                state['step'] += 1
                buf = state['momentum_buffer']
                buf.mul_(momentum).add_(grad, alpha=1 - momentum)
                debiased = buf / (1 - momentum ** state['step'])

                shape = p.shape
                if p.dim() == 0:
                    g2d = debiased.reshape(1, 1)
                elif p.dim() == 1:
                    g2d = debiased.reshape(1, -1)
                elif p.dim() == 2:
                    g2d = debiased
                elif group['flatten_conv']:
                    g2d = debiased.reshape(shape[0], -1)
                else:
                    g2d = debiased.reshape(-1, shape[-1])
                g2d = g2d.to(precond_dtype)

                if 'ql' not in state:
                    state['ql'], state['qr'] = self._init_factors(g2d, group, precond_dtype)

                if do_update:
                    self._update_precond(state['ql'], state['qr'], g2d, group['preconditioner_lr'])

                update = self._precondition(state['ql'], state['qr'], g2d)
                # RMS-clip the preconditioned update for stability (PSGD heuristic)
                rms = update.square().mean().sqrt().clamp(min=1e-30)
                update = update / (rms / 1.1).clamp(min=1.0)
                update = update.reshape(shape).to(p.dtype)

                if group['caution']:
                    mask = (update * grad > 0).to(grad.dtype)
                    mask.div_(mask.mean().clamp_(min=1e-3))
                    update = update * mask
                if group['weight_decay'] != 0 and p.dim() >= 2:
                    update = update.add(p, alpha=group['weight_decay'])

                p.add_(update, alpha=-group['lr'])

        return loss
