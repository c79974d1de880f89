"""Tracing-compatible helpers (reference `timm/layers/trace_utils.py`):
`_assert` stays symbolic under torch.fx instead of branching."""
import torch

try:
    from torch import _assert
except ImportError:
    def _assert(condition: bool, message: str):
        assert condition, message


def _float_to_int(x: float) -> int:
    return int(x)
