"""Deprecated alias module (reference `timm/models/helpers.py`): everything
here moved into `timm_amd.models`; kept so old import paths keep working."""
import warnings

from ._builder import *   # noqa: F401,F403
from ._helpers import *   # noqa: F401,F403
from ._manipulate import *  # noqa: F401,F403
from ._prune import *     # noqa: F401,F403

warnings.warn(
    f'Importing from {__name__} is deprecated, please import via timm_amd.models',
    FutureWarning,
)
