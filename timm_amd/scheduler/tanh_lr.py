"""Tanh LR schedule — implementation lives in the shared core (core.py)."""
from .core import TanhLRScheduler

__all__ = ['TanhLRScheduler']
