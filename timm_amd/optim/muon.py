"""Muon — MomentUm Orthogonalized by Newton-Schulz (reference `timm/optim/muon.py`, 1,056 LoC).

MI355X mapping: the 5-step Newton-Schulz iteration runs entirely in bf16 on
MFMA through hipBLASLt (torch.matmul), with preallocated ping-pong buffers
(the reference's addmm/baddbmm scheme, `muon.py:183-197`).  1-d params and
unsuitable tensors fall back to AdamW-style updates inside the same
optimizer (reference `muon.py:650`).
"""
import math
import os
from typing import List, Optional, Tuple

import torch
from torch import Tensor
from torch.optim.optimizer import Optimizer

from .. import ops

# quintic Newton-Schulz coefficients (Keller Jordan's tuned set; reference `muon.py:46-84`)
NS_COEFFS = (3.4445, -4.7750, 2.0315)


def zeropower_via_newtonschulz(
        G: Tensor,
        steps: int = 5,
        coeffs: Tuple[float, float, float] = NS_COEFFS,
        eps: float = 1e-7,
) -> Tensor:
    """Orthogonalize G via quintic Newton-Schulz iteration in bf16.

    Computes an approximation of UV^T (from the SVD G = USV^T).  All matmuls
    are bf16 → MFMA path.  Works batched over leading dims.
    Reference `muon.py:118-202`.
    """
    assert G.ndim >= 2
    a, b, c = coeffs
    X = G.to(torch.bfloat16)
    transposed = False
    if G.size(-2) > G.size(-1):
        X = X.mT
        transposed = True

    # normalize so top singular value <= 1
    X = X / (X.norm(dim=(-2, -1), keepdim=True) + eps)

    use_kernels = X.is_cuda and os.environ.get('TIMM_AMD_MUON_NS', 'hip') != 'torch'
    ext = ops._load_extension() if use_kernels else None
    if ext is not None:
        # in-house batched MFMA kernels (muon_ns.hip): A and B are symmetric,
        # so every product runs as row-major NT except the final BX (NN with
        # transposed staging).  fp32 accumulate, bf16 IO.
        shape = X.shape
        Xb = X.reshape(-1, shape[-2], shape[-1]).contiguous()
        for _ in range(steps):
            A = ext.ns_gemm_nt(Xb, Xb, None, 1.0, 0.0)
            B = ext.ns_gemm_nt(A, A, A, c, b)
            Xb = ext.ns_gemm_nn(B, Xb, Xb, 1.0, a)
        X = Xb.reshape(shape)
    else:
        for _ in range(steps):
            A = X @ X.mT
            B = b * A + c * (A @ A)
            X = a * X + B @ X

    if transposed:
        X = X.mT
    return X.to(G.dtype)


def _lr_scale(shape, mode: str = 'match_rms_adamw') -> float:
    """Per-param LR scaling rules (reference `muon.py:205-260`)."""
    A, B = shape[-2], shape[-1]
    if mode == 'match_rms_adamw':
        # scale so update RMS matches AdamW's typical 0.2-0.4 range
        return 0.2 * math.sqrt(max(A, B))
    if mode == 'spectral':
        return math.sqrt(max(1., A / B))
    if mode == 'shape':
        return max(1., A / B) ** 0.5
    return 1.0


def _muon_suitable(p: Tensor) -> bool:
    """Heuristics for which params take the Muon path (reference `muon.py:262-330`):
    2d+ weight matrices; embeddings/heads and 1d params go to the AdamW path."""
    return p.ndim >= 2


class Muon(Optimizer):
    """Muon optimizer w/ internal AdamW fallback for non-matrix params
    (reference `muon.py:650`)."""

    def __init__(
            self,
            params,
            lr: float = 0.02,
            momentum: float = 0.95,
            nesterov: bool = True,
            ns_steps: int = 5,
            weight_decay: float = 0.,
            lr_scale_mode: str = 'match_rms_adamw',
            # AdamW fallback args for 1d / unsuitable params
            adamw_lr: Optional[float] = None,
            adamw_betas: Tuple[float, float] = (0.9, 0.95),
            adamw_eps: float = 1e-8,
            flatten_conv: bool = True,
            caution: bool = False,
    ):
        defaults = dict(
            lr=lr, momentum=momentum, nesterov=nesterov, ns_steps=ns_steps,
            weight_decay=weight_decay, lr_scale_mode=lr_scale_mode,
            adamw_lr=adamw_lr if adamw_lr is not None else lr,
            adamw_betas=adamw_betas, adamw_eps=adamw_eps,
            flatten_conv=flatten_conv, caution=caution,
        )
        super().__init__(params, defaults)

    def __setstate__(self, state):
        super().__setstate__(state)
        for group in self.param_groups:
            group.setdefault('nesterov', True)
            group.setdefault('caution', False)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            lr = group['lr']
            momentum = group['momentum']
            nesterov = group['nesterov']
            ns_steps = group['ns_steps']
            wd = group['weight_decay']
            use_muon_group = group.get('use_muon', None)

            adamw_params, adamw_grads, adamw_m, adamw_v = [], [], [], []
            adamw_step = None
            # muon-path work items batched by 2d shape so the NS kernels run
            # one [G,M,N] launch per shape (fills all 256 CUs instead of
            # per-matrix grids of a few dozen workgroups)
            ns_queue = {}  # (M, N) -> list of (param, grad, u2, orig_shape)

            for p in group['params']:
                if p.grad is None:
                    continue
                g = p.grad
                state = self.state[p]

                use_muon = use_muon_group if use_muon_group is not None else _muon_suitable(p)
                if use_muon:
                    if 'momentum_buffer' not in state:
                        state['momentum_buffer'] = torch.zeros_like(g)
                    buf = state['momentum_buffer']
                    buf.lerp_(g, 1 - momentum)
                    u = g.lerp_(buf, momentum) if nesterov else buf

                    shape = u.shape
                    if u.ndim > 2:
                        if group['flatten_conv']:
                            u2 = u.reshape(shape[0], -1)  # conv [O,I,kh,kw] -> [O, I*kh*kw]
                        else:
                            u2 = u.reshape(-1, shape[-1])
                    else:
                        u2 = u
                    ns_queue.setdefault(tuple(u2.shape), []).append((p, g, u2, shape))
                else:
                    if 'exp_avg' not in state:
                        state['exp_avg'] = torch.zeros_like(p, dtype=torch.float32)
                        state['exp_avg_sq'] = torch.zeros_like(p, dtype=torch.float32)
                        state['adamw_step'] = 0
                    state['adamw_step'] += 1
                    adamw_step = state['adamw_step']
                    adamw_params.append(p)
                    adamw_grads.append(g)
                    adamw_m.append(state['exp_avg'])
                    adamw_v.append(state['exp_avg_sq'])

            # run NS batched per shape, then apply updates
            for (M, N), items in ns_queue.items():
                stack = torch.stack([u2 for _p, _g, u2, _s in items]) if len(items) > 1 \
                    else items[0][2].unsqueeze(0)
                ortho = zeropower_via_newtonschulz(stack, steps=ns_steps)
                scale = _lr_scale((M, N), group['lr_scale_mode'])
                for (p, g, _u2, shape), o in zip(items, ortho.unbind(0)):
                    if wd:
                        p.mul_(1 - lr * wd)
                    upd = o.to(p.dtype).reshape(shape)
                    if group['caution']:
                        mask = (upd * g > 0).to(g.dtype)
                        mask.div_(mask.mean().clamp_(min=1e-3))
                        p.add_(upd * mask, alpha=-lr * scale)
                    else:
                        p.add_(upd, alpha=-lr * scale)

            if adamw_params:
                b1, b2 = group['adamw_betas']
                ops.fused_adamw_step(
                    adamw_params, adamw_grads, adamw_m, adamw_v,
                    lr=group['adamw_lr'], beta1=b1, beta2=b2, eps=group['adamw_eps'],
                    weight_decay=wd, step=adamw_step, caution=group['caution'],
                )
        return loss


class AdaMuon(Muon):
    """AdaMuon variant: second-moment normalization applied to the
    orthogonalized update (sign-scale-invariant Adam-style step on top of NS)."""

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            lr = group['lr']
            momentum = group['momentum']
            nesterov = group['nesterov']
            ns_steps = group['ns_steps']
            wd = group['weight_decay']

            adamw_params, adamw_grads, adamw_m, adamw_v = [], [], [], []
            adamw_step = None

            for p in group['params']:
                if p.grad is None:
                    continue
                g = p.grad
                state = self.state[p]
                if _muon_suitable(p):
                    if 'momentum_buffer' not in state:
                        state['momentum_buffer'] = torch.zeros_like(g)
                        state['second_moment'] = torch.zeros_like(g, dtype=torch.float32)
                    buf = state['momentum_buffer']
                    buf.lerp_(g, 1 - momentum)
                    u = g.lerp_(buf, momentum) if nesterov else buf
                    shape = u.shape
                    u2 = u.reshape(shape[0], -1) if u.ndim > 2 else u
                    u2 = zeropower_via_newtonschulz(u2, steps=ns_steps)
                    v = state['second_moment']
                    v.mul_(0.999).addcmul_(u2.float().reshape(v.shape), u2.float().reshape(v.shape), value=0.001)
                    upd = (u2.reshape(shape).float() / (v.sqrt() + 1e-8)).to(p.dtype)
                    scale = _lr_scale(u2.shape, group['lr_scale_mode']) * 0.2
                    if wd:
                        p.mul_(1 - lr * wd)
                    p.add_(upd, alpha=-lr * scale)
                else:
                    if 'exp_avg' not in state:
                        state['exp_avg'] = torch.zeros_like(p, dtype=torch.float32)
                        state['exp_avg_sq'] = torch.zeros_like(p, dtype=torch.float32)
                        state['adamw_step'] = 0
                    state['adamw_step'] += 1
                    adamw_step = state['adamw_step']
                    adamw_params.append(p)
                    adamw_grads.append(g)
                    adamw_m.append(state['exp_avg'])
                    adamw_v.append(state['exp_avg_sq'])

            if adamw_params:
                b1, b2 = group['adamw_betas']
                ops.fused_adamw_step(
                    adamw_params, adamw_grads, adamw_m, adamw_v,
                    lr=group['adamw_lr'], beta1=b1, beta2=b2, eps=group['adamw_eps'],
                    weight_decay=wd, step=adamw_step, caution=False,
                )
        return loss
