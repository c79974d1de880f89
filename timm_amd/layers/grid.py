"""Meshgrid helpers (reference `timm/layers/grid.py`)."""
from typing import Tuple

import torch

__all__ = ['ndgrid', 'meshgrid']


def ndgrid(*tensors) -> Tuple[torch.Tensor, ...]:
    """Matrix-indexed ('ij') meshgrid, explicit about indexing order."""
    return torch.meshgrid(*tensors, indexing='ij')


def meshgrid(*tensors) -> Tuple[torch.Tensor, ...]:
    """Cartesian-indexed ('xy') meshgrid."""
    return torch.meshgrid(*tensors, indexing='xy')
