"""Deprecated import location — use timm_amd.models (reference `timm/models/fx_features.py`)."""
from ._features_fx import *

import warnings
warnings.warn(f"Importing from {__name__} is deprecated, please import via timm_amd.models", FutureWarning)
