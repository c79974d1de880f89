"""Polynomial LR schedule — implementation lives in the shared core (core.py)."""
from .core import PolyLRScheduler

__all__ = ['PolyLRScheduler']
