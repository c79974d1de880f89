"""Deprecated import location — use timm_amd.models (reference `timm/models/factory.py`)."""
from ._factory import *

import warnings
warnings.warn(f"Importing from {__name__} is deprecated, please import via timm_amd.models", FutureWarning)
