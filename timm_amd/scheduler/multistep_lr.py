"""Multi-step LR schedule — implementation lives in the shared core (core.py)."""
from .core import MultiStepLRScheduler

__all__ = ['MultiStepLRScheduler']
