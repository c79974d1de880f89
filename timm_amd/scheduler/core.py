"""LR scheduling core — single-module redesign of the reference scheduler zoo.

Behavioral parity targets (reference files, /root/reference/timm/scheduler/):
  scheduler.py:8-127 (explicit-timestep base + seeded noise), cosine_lr.py:19
  (SGDR restarts + k-decay), tanh_lr.py, poly_lr.py, step_lr.py,
  multistep_lr.py, plateau_lr.py.

Architecture here is deliberately different from the reference's
one-class-per-file `_get_lr` pattern: all annealing schedules share one
`CyclicLRScheduler` driver (warmup ramp -> SGDR cycle geometry -> per-schedule
*shape function* mapping progress in [0,1] to the remaining lr fraction), and
the two stepped schedules share a `SteppedLRScheduler` driver with a
decay-count hook.  The schedule laws themselves are pure functions, so the
value curves match the reference exactly while the machinery is written once.

Timestep convention (same as reference): `step(epoch)` is called at the end
of each epoch with the *next* epoch index; `step_update(num_updates)` after
each optimizer update when scheduling per-update.  No hidden `last_epoch`.
"""
import bisect
import math
from typing import Any, Dict, List, Optional, Sequence, Union

import torch

__all__ = [
    'Scheduler', 'CyclicLRScheduler', 'SteppedLRScheduler',
    'CosineLRScheduler', 'TanhLRScheduler', 'PolyLRScheduler',
    'StepLRScheduler', 'MultiStepLRScheduler', 'PlateauLRScheduler',
]


def sgdr_cycle(t: float, span0: float, mul: float):
    """Resolve SGDR cycle geometry at time ``t``.

    Returns (cycle_index, time_within_cycle, cycle_span).  With ``mul != 1``
    cycle i has span ``span0 * mul**i`` and starts at the geometric partial sum
    ``span0 * (1 - mul**i) / (1 - mul)``.
    """
    if mul == 1.0:
        i = int(t // span0)
        return i, t - i * span0, span0
    i = int(math.floor(math.log(1 - t / span0 * (1 - mul), mul)))
    start = span0 * (1 - mul ** i) / (1 - mul)
    return i, t - start, span0 * mul ** i


def sgdr_total_span(span0: float, mul: float, cycles: int) -> int:
    """Total timesteps covered by ``cycles`` SGDR cycles."""
    if mul == 1.0:
        return int(span0 * cycles)
    return int(math.floor(-span0 * (mul ** cycles - 1) / (1 - mul)))


class Scheduler:
    """Explicit-timestep param-group scheduler base.

    Subclasses implement ``schedule(t) -> List[float]``.  The base owns group
    discovery, per-group ``lr_scale``, and the seeded noise overlay.
    """

    def __init__(
            self,
            optimizer: torch.optim.Optimizer,
            param_group_field: str = 'lr',
            t_in_epochs: bool = True,
            noise_range_t: Union[None, float, Sequence[float]] = None,
            noise_type: str = 'normal',
            noise_pct: float = 0.67,
            noise_std: float = 1.0,
            noise_seed: Optional[int] = None,
            initialize: bool = True,
    ):
        self.optimizer = optimizer
        self.param_group_field = param_group_field
        self._initial_param_group_field = 'initial_' + param_group_field
        src, dst = (param_group_field, self._initial_param_group_field)
        if not initialize:
            src, dst = dst, src  # expect snapshots already recorded
        for idx, group in enumerate(self.optimizer.param_groups):
            if src not in group:
                raise KeyError(f'{src} missing from param_groups[{idx}]')
            if initialize:
                group.setdefault(dst, group[src])
        self.base_values = [
            g[self._initial_param_group_field] for g in self.optimizer.param_groups]
        self.t_in_epochs = t_in_epochs
        self.metric = None
        self.noise_range_t = noise_range_t
        self.noise_type = noise_type
        self.noise_pct = noise_pct
        self.noise_std = noise_std
        self.noise_seed = 42 if noise_seed is None else noise_seed
        self.update_groups(self.base_values)

    # -- subclass hook ------------------------------------------------------
    def schedule(self, t: int) -> List[float]:
        raise NotImplementedError

    # reference-API alias (some external code calls _get_lr directly)
    def _get_lr(self, t: int) -> List[float]:
        return self.schedule(t)

    # -- public stepping API ------------------------------------------------
    def step(self, epoch: int, metric: Optional[float] = None) -> None:
        self.metric = metric
        if self.t_in_epochs:
            self._apply(epoch)

    def step_update(self, num_updates: int, metric: Optional[float] = None) -> None:
        self.metric = metric
        if not self.t_in_epochs:
            self._apply(num_updates)

    def _apply(self, t: int) -> None:
        values = self.schedule(t)
        noise = self._noise_at(t)
        if noise is not None:
            values = [v + v * noise for v in values]
        self.update_groups(values)

    def update_groups(self, values) -> None:
        if not isinstance(values, (list, tuple)):
            values = [values] * len(self.optimizer.param_groups)
        for group, value in zip(self.optimizer.param_groups, values):
            scale = group.get('lr_scale', None)
            group[self.param_group_field] = value * scale if scale is not None else value

    # -- noise overlay ------------------------------------------------------
    def _noise_at(self, t: int) -> Optional[float]:
        rng_t = self.noise_range_t
        if rng_t is None:
            return None
        if isinstance(rng_t, (list, tuple)):
            active = rng_t[0] <= t < rng_t[1]
        else:
            active = t >= rng_t
        if not active:
            return None
        g = torch.Generator().manual_seed(self.noise_seed + t)
        if self.noise_type == 'normal':
            # rejection-sample a bounded gaussian so a tail draw can't spike lr
            while True:
                draw = torch.randn(1, generator=g).item()
                if abs(draw) < self.noise_pct:
                    return draw
        return 2 * (torch.rand(1, generator=g).item() - 0.5) * self.noise_pct

    # kept for subclass/plateau reuse
    def _is_apply_noise(self, t: int) -> bool:
        rng_t = self.noise_range_t
        if rng_t is None:
            return False
        if isinstance(rng_t, (list, tuple)):
            return rng_t[0] <= t < rng_t[1]
        return t >= rng_t

    # -- state --------------------------------------------------------------
    def state_dict(self) -> Dict[str, Any]:
        return {k: v for k, v in self.__dict__.items() if k != 'optimizer'}

    def load_state_dict(self, state_dict: Dict[str, Any]) -> None:
        self.__dict__.update(state_dict)


class CyclicLRScheduler(Scheduler):
    """Warmup ramp + SGDR cycles + a per-schedule shape function.

    ``_shape(x)`` maps decay progress ``x`` (``(t_curr/t_span) ** k``) to the
    fraction of ``(peak - lr_min)`` remaining; peak of cycle i is
    ``base * cycle_decay**i``.  Past ``cycle_limit`` cycles lr pins to lr_min.
    """

    #: when True, warmup ramps toward the schedule value at t=warmup_t rather
    #: than the base lr (the reference's tanh scheduler behaves this way)
    _ramp_to_schedule = False

    def __init__(
            self,
            optimizer: torch.optim.Optimizer,
            t_initial: int,
            lr_min: float = 0.,
            cycle_mul: float = 1.,
            cycle_decay: float = 1.,
            cycle_limit: int = 1,
            warmup_t: int = 0,
            warmup_lr_init: float = 0,
            warmup_prefix: bool = False,
            t_in_epochs: bool = True,
            noise_range_t=None,
            noise_pct: float = 0.67,
            noise_std: float = 1.0,
            noise_seed: int = 42,
            k_decay: float = 1.0,
            initialize: bool = True,
    ):
        super().__init__(
            optimizer, 'lr',
            t_in_epochs=t_in_epochs,
            noise_range_t=noise_range_t,
            noise_pct=noise_pct,
            noise_std=noise_std,
            noise_seed=noise_seed,
            initialize=initialize,
        )
        assert t_initial > 0 and lr_min >= 0
        self.t_initial = t_initial
        self.lr_min = lr_min
        self.cycle_mul = cycle_mul
        self.cycle_decay = cycle_decay
        self.cycle_limit = cycle_limit
        self.warmup_t = warmup_t
        self.warmup_lr_init = warmup_lr_init
        self.warmup_prefix = warmup_prefix
        self.k_decay = k_decay
        self.warmup_steps = self._warmup_slopes()
        if warmup_t:
            self.update_groups(self.warmup_lr_init)

    def _warmup_slopes(self) -> List[float]:
        if not self.warmup_t:
            return [1 for _ in self.base_values]
        if self._ramp_to_schedule and not self.warmup_prefix:
            targets = self._anneal(self.warmup_t)
        else:
            targets = self.base_values
        return [(v - self.warmup_lr_init) / self.warmup_t for v in targets]

    def _shape(self, x: float) -> float:
        raise NotImplementedError

    def _anneal(self, t: float) -> List[float]:
        if self.warmup_prefix:
            t = t - self.warmup_t
        i, t_curr, t_span = sgdr_cycle(t, self.t_initial, self.cycle_mul)
        if i >= self.cycle_limit:
            return [self.lr_min for _ in self.base_values]
        frac = self._shape((t_curr / t_span) ** self.k_decay)
        peak_scale = self.cycle_decay ** i
        return [
            self.lr_min + (base * peak_scale - self.lr_min) * frac
            for base in self.base_values
        ]

    def schedule(self, t: int) -> List[float]:
        if t < self.warmup_t:
            return [self.warmup_lr_init + t * s for s in self.warmup_steps]
        return self._anneal(t)

    def get_cycle_length(self, cycles: int = 0) -> int:
        cycles = max(1, cycles or self.cycle_limit)
        total = sgdr_total_span(self.t_initial, self.cycle_mul, cycles)
        return total + self.warmup_t if self.warmup_prefix else total


class CosineLRScheduler(CyclicLRScheduler):
    """Cosine annealing w/ SGDR restarts (arxiv 1608.03983) and k-decay
    (arxiv 2004.05909).  Reference: timm/scheduler/cosine_lr.py:19."""

    def _shape(self, x: float) -> float:
        return 0.5 * (1 + math.cos(math.pi * x))


class PolyLRScheduler(CyclicLRScheduler):
    """Polynomial decay ``(1 - x)**power`` w/ warmup/cycles/k-decay.
    Reference: timm/scheduler/poly_lr.py."""

    def __init__(self, optimizer, t_initial, power: float = 0.5, **kwargs):
        self.power = power
        super().__init__(optimizer, t_initial, **kwargs)

    def _shape(self, x: float) -> float:
        return (1 - x) ** self.power


class TanhLRScheduler(CyclicLRScheduler):
    """Hyperbolic-tangent decay (arxiv 1806.01593): the shape sweeps
    ``0.5*(1 - tanh(.))`` from lb to ub.  Reference: timm/scheduler/tanh_lr.py.
    Warmup ramps to the schedule's value at warmup end (not the base lr)."""

    _ramp_to_schedule = True

    def __init__(self, optimizer, t_initial, lb: float = -7., ub: float = 3., **kwargs):
        assert lb < ub
        self.lb = lb
        self.ub = ub
        kwargs.pop('k_decay', None)  # tanh has no k-decay in the reference
        super().__init__(optimizer, t_initial, **kwargs)

    def _shape(self, x: float) -> float:
        return 0.5 * (1 - math.tanh(self.lb * (1. - x) + self.ub * x))


class SteppedLRScheduler(Scheduler):
    """Warmup ramp + multiplicative decay ``base * rate**count(t)``.

    Subclasses supply ``_decay_count(t)``.  Unlike the cyclic schedules,
    warmup_prefix defaults True (decay clock starts after warmup)."""

    def __init__(
            self,
            optimizer: torch.optim.Optimizer,
            decay_t,
            decay_rate: float = 1.,
            warmup_t: int = 0,
            warmup_lr_init: float = 0,
            warmup_prefix: bool = True,
            t_in_epochs: bool = True,
            noise_range_t=None,
            noise_pct: float = 0.67,
            noise_std: float = 1.0,
            noise_seed: int = 42,
            initialize: bool = True,
    ):
        super().__init__(
            optimizer, 'lr',
            t_in_epochs=t_in_epochs,
            noise_range_t=noise_range_t,
            noise_pct=noise_pct,
            noise_std=noise_std,
            noise_seed=noise_seed,
            initialize=initialize,
        )
        self.decay_t = decay_t
        self.decay_rate = decay_rate
        self.warmup_t = warmup_t
        self.warmup_lr_init = warmup_lr_init
        self.warmup_prefix = warmup_prefix
        if warmup_t:
            self.warmup_steps = [(v - warmup_lr_init) / warmup_t for v in self.base_values]
            self.update_groups(warmup_lr_init)
        else:
            self.warmup_steps = [1 for _ in self.base_values]

    def _decay_count(self, t: int) -> int:
        raise NotImplementedError

    def schedule(self, t: int) -> List[float]:
        if t < self.warmup_t:
            return [self.warmup_lr_init + t * s for s in self.warmup_steps]
        if self.warmup_prefix:
            t = t - self.warmup_t
        factor = self.decay_rate ** self._decay_count(t)
        return [v * factor for v in self.base_values]


class StepLRScheduler(SteppedLRScheduler):
    """Fixed-interval step decay.  Reference: timm/scheduler/step_lr.py."""

    def _decay_count(self, t: int) -> int:
        return int(t // self.decay_t)


class MultiStepLRScheduler(SteppedLRScheduler):
    """Milestone step decay.  Reference: timm/scheduler/multistep_lr.py."""

    def _decay_count(self, t: int) -> int:
        # milestone epoch M takes effect at the step() call made at the end of
        # epoch M-1 (which passes t=M-1+1 semantics via t+1 below)
        return bisect.bisect_right(self.decay_t, t + 1)

    # reference-API alias
    def get_curr_decay_steps(self, t: int) -> int:
        return self._decay_count(t)


class PlateauLRScheduler(Scheduler):
    """Metric-plateau decay: wraps torch ReduceLROnPlateau behind the explicit
    timestep API, with warmup and noise handled here.
    Reference: timm/scheduler/plateau_lr.py."""

    def __init__(
            self,
            optimizer,
            decay_rate: float = 0.1,
            patience_t: int = 10,
            verbose: bool = True,
            threshold: float = 1e-4,
            cooldown_t: int = 0,
            warmup_t: int = 0,
            warmup_lr_init: float = 0,
            lr_min: float = 0,
            mode: str = 'max',
            noise_range_t=None,
            noise_type='normal',
            noise_pct: float = 0.67,
            noise_std: float = 1.0,
            noise_seed: Optional[int] = None,
            initialize: bool = True,
    ):
        super().__init__(
            optimizer, 'lr',
            noise_range_t=noise_range_t,
            noise_type=noise_type,
            noise_pct=noise_pct,
            noise_std=noise_std,
            noise_seed=noise_seed,
            initialize=initialize,
        )
        self.inner = torch.optim.lr_scheduler.ReduceLROnPlateau(
            self.optimizer,
            mode=mode,
            factor=decay_rate,
            patience=patience_t,
            threshold=threshold,
            cooldown=cooldown_t,
            min_lr=lr_min,
        )
        self.warmup_t = warmup_t
        self.warmup_lr_init = warmup_lr_init
        if warmup_t:
            self.warmup_steps = [(v - warmup_lr_init) / warmup_t for v in self.base_values]
            self.update_groups(warmup_lr_init)
        else:
            self.warmup_steps = [1 for _ in self.base_values]
        self._pre_noise_lrs = None

    # reference attribute name for external pokes
    @property
    def lr_scheduler(self):
        return self.inner

    def schedule(self, t: int) -> List[float]:
        raise AssertionError('PlateauLRScheduler drives groups via step() only')

    def step(self, epoch: int, metric: Optional[float] = None) -> None:
        if epoch <= self.warmup_t:
            self.update_groups(
                [self.warmup_lr_init + epoch * s for s in self.warmup_steps])
            return
        if self._pre_noise_lrs is not None:
            # strip the previous noise perturbation before the plateau logic
            # sees (and potentially decays) the lr
            for group, lr in zip(self.optimizer.param_groups, self._pre_noise_lrs):
                group['lr'] = lr
            self._pre_noise_lrs = None
        self.inner.step(metric, epoch)
        if self._is_apply_noise(epoch):
            noise = self._noise_at(epoch)
            saved = []
            for group in self.optimizer.param_groups:
                lr = float(group['lr'])
                saved.append(lr)
                group['lr'] = lr * (1 + noise)
            self._pre_noise_lrs = saved

    def step_update(self, num_updates: int, metric: Optional[float] = None) -> None:
        return None

    def state_dict(self):
        return {'best': self.inner.best, 'last_epoch': self.inner.last_epoch}

    def load_state_dict(self, state_dict):
        self.inner.best = state_dict['best']
        if 'last_epoch' in state_dict:
            self.inner.last_epoch = state_dict['last_epoch']
