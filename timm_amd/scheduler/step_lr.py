"""Step LR schedule — implementation lives in the shared core (core.py)."""
from .core import StepLRScheduler

__all__ = ['StepLRScheduler']
