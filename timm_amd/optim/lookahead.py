"""Lookahead wrapper: k fast steps, then pull slow weights toward the fast
ones (reference `timm/optim/lookahead.py`; paper arxiv 1907.08610).

Slow-weight buffers live inside the wrapped optimizer's per-param state so a
single state_dict round-trips both optimizers.
"""
from collections import OrderedDict, defaultdict
from typing import Callable, Dict

import torch
from torch.optim.optimizer import Optimizer


class Lookahead(Optimizer):
    def __init__(self, base_optimizer, alpha=0.5, k=6):
        if not 0.0 <= alpha <= 1.0:
            raise ValueError(f'Invalid slow update rate: {alpha}')
        if k < 1:
            raise ValueError(f'Invalid lookahead steps: {k}')
        # Deliberately skip Optimizer.__init__: this wrapper aliases the base
        # optimizer's groups/defaults instead of owning its own.
        self._optimizer_step_pre_hooks: Dict[int, Callable] = OrderedDict()
        self._optimizer_step_post_hooks: Dict[int, Callable] = OrderedDict()
        extra = dict(lookahead_alpha=alpha, lookahead_k=k, lookahead_step=0)
        self._base_optimizer = base_optimizer
        self.param_groups = base_optimizer.param_groups
        self.defaults = base_optimizer.defaults
        self.defaults.update(extra)
        self.state = defaultdict(dict)
        for group in self._base_optimizer.param_groups:
            for name, value in extra.items():
                group.setdefault(name, value)

    @torch.no_grad()
    def update_slow(self, group):
        alpha = group['lookahead_alpha']
        for fast in group['params']:
            if fast.grad is None:
                continue
            state = self._base_optimizer.state[fast]
            slow = state.get('lookahead_slow_buff')
            if slow is None:
                slow = state['lookahead_slow_buff'] = torch.empty_like(fast)
                slow.copy_(fast)
            slow.add_(fast - slow, alpha=alpha)
            fast.copy_(slow)

    def sync_lookahead(self):
        for group in self._base_optimizer.param_groups:
            self.update_slow(group)

    @torch.no_grad()
    def step(self, closure=None):
        loss = self._base_optimizer.step(closure)
        for group in self._base_optimizer.param_groups:
            group['lookahead_step'] += 1
            if group['lookahead_step'] % group['lookahead_k'] == 0:
                self.update_slow(group)
        return loss

    def state_dict(self):
        return self._base_optimizer.state_dict()

    def load_state_dict(self, state_dict):
        self._base_optimizer.load_state_dict(state_dict)
        self.param_groups = self._base_optimizer.param_groups
