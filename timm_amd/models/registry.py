"""Deprecated import location — use timm_amd.models (reference `timm/models/registry.py`)."""
from ._registry import *

import warnings
warnings.warn(f"Importing from {__name__} is deprecated, please import via timm_amd.models", FutureWarning)
