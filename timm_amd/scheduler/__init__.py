"""LR schedulers.

All schedules share the step-driven `Scheduler` base in `core.py`; the
per-name modules are thin shims kept for import parity with the reference.
"""
from .scheduler import Scheduler
from .scheduler_factory import create_scheduler, create_scheduler_v2, scheduler_kwargs

from .cosine_lr import CosineLRScheduler
from .multistep_lr import MultiStepLRScheduler
from .plateau_lr import PlateauLRScheduler
from .poly_lr import PolyLRScheduler
from .step_lr import StepLRScheduler
from .tanh_lr import TanhLRScheduler

__all__ = [
    'Scheduler', 'create_scheduler', 'create_scheduler_v2', 'scheduler_kwargs',
    'CosineLRScheduler', 'MultiStepLRScheduler', 'PlateauLRScheduler',
    'PolyLRScheduler', 'StepLRScheduler', 'TanhLRScheduler',
]
