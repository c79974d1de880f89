"""Scheduler construction from training config.

Behavioral parity: /root/reference/timm/scheduler/scheduler_factory.py:63
(create_scheduler_v2 semantics: epoch->update conversion, noise range scaling,
cycle-length-aware epoch recount).  Structure here is table-driven: each
schedule name maps to (class, set of extra arg-bundles it accepts).
"""
from typing import List, Optional, Union

from torch.optim import Optimizer

from .core import (
    CosineLRScheduler,
    MultiStepLRScheduler,
    PlateauLRScheduler,
    PolyLRScheduler,
    StepLRScheduler,
    TanhLRScheduler,
)

__all__ = ['scheduler_kwargs', 'create_scheduler', 'create_scheduler_v2']

# schedule name -> (class, takes_cycle_args, takes_k_decay)
_SCHEDULES = {
    'cosine': (CosineLRScheduler, True, True),
    'tanh': (TanhLRScheduler, True, False),
    'poly': (PolyLRScheduler, True, True),
    'step': (StepLRScheduler, False, False),
    'multistep': (MultiStepLRScheduler, False, False),
    'plateau': (PlateauLRScheduler, False, False),
}


def scheduler_kwargs(cfg, decreasing_metric: Optional[bool] = None) -> dict:
    """Flatten an argparse/config namespace into create_scheduler_v2 kwargs."""
    get = lambda name, default: getattr(cfg, name, default)  # noqa: E731
    if decreasing_metric is None:
        decreasing_metric = 'loss' in get('eval_metric', 'top1')
    return dict(
        sched=cfg.sched,
        num_epochs=get('epochs', 100),
        decay_epochs=get('decay_epochs', 30),
        decay_milestones=get('decay_milestones', [30, 60]),
        warmup_epochs=get('warmup_epochs', 5),
        cooldown_epochs=get('cooldown_epochs', 0),
        patience_epochs=get('patience_epochs', 10),
        decay_rate=get('decay_rate', 0.1),
        min_lr=get('min_lr', 0.),
        warmup_lr=get('warmup_lr', 1e-5),
        warmup_prefix=get('warmup_prefix', False),
        noise=get('lr_noise', None),
        noise_pct=get('lr_noise_pct', 0.67),
        noise_std=get('lr_noise_std', 1.),
        noise_seed=get('seed', 42),
        cycle_mul=get('lr_cycle_mul', 1.),
        cycle_decay=get('lr_cycle_decay', 0.1),
        cycle_limit=get('lr_cycle_limit', 1),
        k_decay=get('lr_k_decay', 1.),
        plateau_mode='min' if decreasing_metric else 'max',
        step_on_epochs=not get('sched_on_updates', False),
    )


def create_scheduler(args, optimizer: Optimizer, updates_per_epoch: int = 0):
    return create_scheduler_v2(
        optimizer=optimizer,
        **scheduler_kwargs(args),
        updates_per_epoch=updates_per_epoch,
    )


def create_scheduler_v2(
        optimizer: Optimizer,
        sched: str = 'cosine',
        num_epochs: int = 300,
        decay_epochs: int = 90,
        decay_milestones: List[int] = (90, 180, 270),
        cooldown_epochs: int = 0,
        patience_epochs: int = 10,
        decay_rate: float = 0.1,
        min_lr: float = 0,
        warmup_lr: float = 1e-5,
        warmup_epochs: int = 0,
        warmup_prefix: bool = False,
        noise: Union[None, float, List[float]] = None,
        noise_pct: float = 0.67,
        noise_std: float = 1.,
        noise_seed: int = 42,
        cycle_mul: float = 1.,
        cycle_decay: float = 0.1,
        cycle_limit: int = 1,
        k_decay: float = 1.0,
        plateau_mode: str = 'max',
        step_on_epochs: bool = True,
        updates_per_epoch: int = 0,
):
    """Build a scheduler + the (possibly cycle-extended) epoch count.

    When ``step_on_epochs`` is False every epoch-denominated knob is converted
    to optimizer-update units using ``updates_per_epoch``.
    """
    if sched not in _SCHEDULES:
        # reference behavior: unknown/'none' name -> no scheduler
        return None, num_epochs
    cls, takes_cycle, takes_k = _SCHEDULES[sched]

    # convert epoch-denominated quantities to timestep units
    tick = 1 if step_on_epochs else updates_per_epoch
    if not step_on_epochs:
        assert updates_per_epoch > 0, \
            'updates_per_epoch must be set to number of dataloader batches'
    total_t = num_epochs * tick
    warmup_t = warmup_epochs * tick
    cooldown_t = cooldown_epochs * tick

    kwargs = dict(
        warmup_t=warmup_t,
        warmup_lr_init=warmup_lr,
        t_in_epochs=step_on_epochs,
        noise_pct=noise_pct,
        noise_std=noise_std,
        noise_seed=noise_seed,
    )
    # noise window is specified as fraction(s) of the total schedule span
    if noise is not None:
        if isinstance(noise, (list, tuple)):
            rng = [n * total_t for n in noise]
            kwargs['noise_range_t'] = rng[0] if len(rng) == 1 else rng
        else:
            kwargs['noise_range_t'] = noise * total_t
    else:
        kwargs['noise_range_t'] = None

    if sched == 'plateau':
        assert step_on_epochs, 'Plateau LR only supports step per epoch.'
        kwargs.update(dict(
            decay_rate=decay_rate,
            patience_t=patience_epochs,
            cooldown_t=0,
            mode=plateau_mode,
            lr_min=min_lr,
        ))
    elif sched == 'step':
        # stepped schedules keep their class default warmup_prefix=True
        kwargs.update(dict(
            decay_t=decay_epochs * tick,
            decay_rate=decay_rate,
        ))
    elif sched == 'multistep':
        kwargs.update(dict(
            decay_t=[m * tick for m in decay_milestones],
            decay_rate=decay_rate,
        ))
    else:  # cyclic family
        kwargs.update(dict(
            t_initial=total_t,
            lr_min=min_lr,
            warmup_prefix=warmup_prefix,
            cycle_mul=cycle_mul,
            cycle_decay=cycle_decay,
            cycle_limit=cycle_limit,
        ))
        if takes_k:
            kwargs['k_decay'] = k_decay
        if sched == 'poly':
            kwargs['power'] = decay_rate  # decay_rate doubles as the poly power

    lr_scheduler = cls(optimizer, **kwargs)

    # cycle schedulers can extend the run: recompute the epoch budget
    if hasattr(lr_scheduler, 'get_cycle_length'):
        total = lr_scheduler.get_cycle_length() + cooldown_t
        num_epochs = total if step_on_epochs else total // updates_per_epoch

    return lr_scheduler, num_epochs
