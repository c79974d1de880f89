"""Deprecated import location — use timm_amd.models (reference `timm/models/hub.py`)."""
from ._hub import *

import warnings
warnings.warn(f"Importing from {__name__} is deprecated, please import via timm_amd.models", FutureWarning)
