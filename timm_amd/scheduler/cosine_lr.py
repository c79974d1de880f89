"""Cosine LR schedule — implementation lives in the shared core (core.py)."""
from .core import CosineLRScheduler

__all__ = ['CosineLRScheduler']
