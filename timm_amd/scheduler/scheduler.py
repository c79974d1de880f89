"""Scheduler base class — implementation lives in the shared core (core.py)."""
from .core import Scheduler

__all__ = ['Scheduler']
