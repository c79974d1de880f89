"""AdaHessian (arxiv 2006.00719): second-order optimizer using a Hutchinson
estimate of the Hessian diagonal as the Adam second moment.

Behavioral parity: /root/reference/timm/optim/adahessian.py.  Exposes
``is_second_order = True`` — the train loop must call ``loss.backward(
create_graph=True)`` so the Hessian-vector products can be taken.
"""
import torch
from torch.optim.optimizer import Optimizer

__all__ = ['Adahessian']


class Adahessian(Optimizer):
    """AdaHessian: Adam with sqrt of an EMA of the squared Hessian diagonal
    (optionally spatially averaged for conv kernels) as the denominator."""

    def __init__(
            self,
            params,
            lr=0.1,
            betas=(0.9, 0.999),
            eps=1e-8,
            weight_decay=0.0,
            hessian_power=1.0,
            update_each=1,
            n_samples=1,
            avg_conv_kernel=False,
    ):
        if not 0.0 <= hessian_power <= 1.0:
            raise ValueError(f'Invalid Hessian power value: {hessian_power}')
        self.n_samples = n_samples
        self.update_each = update_each
        self.avg_conv_kernel = avg_conv_kernel
        # one generator per device lazily; deterministic Rademacher draws
        self.seed = 2147483647
        self.generator = torch.Generator().manual_seed(self.seed)

        defaults = dict(
            lr=lr, betas=betas, eps=eps, weight_decay=weight_decay,
            hessian_power=hessian_power)
        super().__init__(params, defaults)

        for p in self.get_params():
            p.hess = 0.0
            self.state[p]['hessian step'] = 0

    @property
    def is_second_order(self):
        return True

    def get_params(self):
        """All trainable params across groups."""
        return (p for group in self.param_groups for p in group['params'] if p.requires_grad)

    def zero_hessian(self):
        for p in self.get_params():
            if not isinstance(p.hess, float) and self.state[p]['hessian step'] % self.update_each == 0:
                p.hess.zero_()

    @torch.no_grad()
    def set_hessian(self):
        """Hutchinson estimate: E[z * H z] over Rademacher z equals diag(H)."""
        params = []
        for p in filter(lambda p: p.grad is not None, self.get_params()):
            if self.state[p]['hessian step'] % self.update_each == 0:
                params.append(p)
            self.state[p]['hessian step'] += 1
        if len(params) == 0:
            return

        if self.generator.device != params[0].device:
            self.generator = torch.Generator(params[0].device).manual_seed(self.seed)

        grads = [p.grad for p in params]
        for i in range(self.n_samples):
            zs = [
                torch.randint(0, 2, p.size(), generator=self.generator, device=p.device,
                              dtype=p.dtype) * 2.0 - 1.0
                for p in params
            ]
            with torch.enable_grad():
                h_zs = torch.autograd.grad(
                    grads, params, grad_outputs=zs,
                    only_inputs=True, retain_graph=i < self.n_samples - 1)
            for h_z, z, p in zip(h_zs, zs, params):
                p.hess += h_z * z / self.n_samples

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        self.zero_hessian()
        self.set_hessian()

        for group in self.param_groups:
            for p in group['params']:
                if p.grad is None or p.hess is None or isinstance(p.hess, float):
                    continue

                hess = p.hess
                if self.avg_conv_kernel and p.dim() == 4:
                    hess = torch.abs(hess).mean(dim=[2, 3], keepdim=True).expand_as(hess).clone()

                # decoupled weight decay
                p.mul_(1 - group['lr'] * group['weight_decay'])

                state = self.state[p]
                if len(state) == 1:  # only 'hessian step' present
                    state['step'] = 0
                    state['exp_avg'] = torch.zeros_like(p)
                    state['exp_hessian_diag_sq'] = torch.zeros_like(p)

                exp_avg, exp_h2 = state['exp_avg'], state['exp_hessian_diag_sq']
                beta1, beta2 = group['betas']
                state['step'] += 1

                exp_avg.mul_(beta1).add_(p.grad, alpha=1 - beta1)
                exp_h2.mul_(beta2).addcmul_(hess, hess, value=1 - beta2)

                bc1 = 1 - beta1 ** state['step']
                bc2 = 1 - beta2 ** state['step']

                k = group['hessian_power']
                denom = (exp_h2 / bc2).pow_(k / 2).add_(group['eps'])
                p.addcdiv_(exp_avg, denom, value=-group['lr'] / bc1)

        return loss
