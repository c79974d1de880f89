"""Plateau LR schedule — implementation lives in the shared core (core.py)."""
from .core import PlateauLRScheduler

__all__ = ['PlateauLRScheduler']
