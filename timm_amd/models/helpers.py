"""Deprecated import location — use timm_amd.models (reference `timm/models/helpers.py`)."""
from ._builder import *
from ._helpers import *
from ._manipulate import *
from ._prune import *

import warnings
warnings.warn(f"Importing from {__name__} is deprecated, please import via timm_amd.models", FutureWarning)
