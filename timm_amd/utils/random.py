"""Global RNG seeding across torch / numpy / python (reference `timm/utils/random.py:6`)."""
import random as _py_random

import numpy as _np
import torch as _torch


def random_seed(seed=42, rank=0):
    """Seed every RNG stream; rank offsets keep distributed workers decorrelated."""
    effective = seed + rank
    _torch.manual_seed(effective)
    _np.random.seed(effective)
    _py_random.seed(effective)
