"""Global layer behaviour flags.

Parity with reference `timm/layers/config.py:101,137` (`set_layer_config`,
`use_fused_attn`).  On this framework `use_fused_attn` gates the hand-written
CDNA4 HIP attention kernel; exportable/scriptable modes force the unfused
PyTorch composition.
"""
import os
import warnings
from contextlib import contextmanager
from typing import Any, Optional

__all__ = [
    'is_no_jit', 'set_no_jit', 'is_exportable', 'set_exportable', 'is_scriptable', 'set_scriptable',
    'set_layer_config', 'use_fused_attn', 'set_fused_attn', 'use_reentrant_ckpt', 'set_reentrant_ckpt',
]

# module-level state
_NO_JIT = False
_EXPORTABLE = False
_SCRIPTABLE = False

# 0 == off, 1 == on (fused HIP kernel when available on device)
_USE_FUSED_ATTN = int(os.environ.get('TIMM_AMD_FUSED_ATTN', os.environ.get('TIMM_FUSED_ATTN', '1')))

_USE_REENTRANT_CKPT = False


def is_no_jit():
    return _NO_JIT


def set_no_jit(mode: bool) -> bool:
    global _NO_JIT
    _NO_JIT = mode
    return True


def is_exportable():
    return _EXPORTABLE


def set_exportable(mode: bool) -> bool:
    global _EXPORTABLE
    _EXPORTABLE = mode
    return True


def is_scriptable():
    return _SCRIPTABLE


def set_scriptable(mode: bool) -> bool:
    global _SCRIPTABLE
    _SCRIPTABLE = mode
    return True


def use_fused_attn(experimental: bool = False) -> bool:
    if _EXPORTABLE or _SCRIPTABLE or _NO_JIT:
        return False
    return _USE_FUSED_ATTN > 0


def set_fused_attn(enable: bool = True, experimental: bool = False):
    global _USE_FUSED_ATTN
    _USE_FUSED_ATTN = 1 if enable else 0


def use_reentrant_ckpt() -> bool:
    return _USE_REENTRANT_CKPT


def set_reentrant_ckpt(enable: bool = True):
    global _USE_REENTRANT_CKPT
    _USE_REENTRANT_CKPT = enable


@contextmanager
def set_layer_config(
        scriptable: Optional[bool] = None,
        exportable: Optional[bool] = None,
        no_jit: Optional[bool] = None,
        no_activation_jit: Optional[bool] = None,
):
    """Layer config context manager (reference `timm/layers/config.py:101`)."""
    global _SCRIPTABLE, _EXPORTABLE, _NO_JIT
    prev = _SCRIPTABLE, _EXPORTABLE, _NO_JIT
    if scriptable is not None:
        _SCRIPTABLE = scriptable
    if exportable is not None:
        _EXPORTABLE = exportable
    if no_jit is not None:
        _NO_JIT = no_jit
    try:
        yield
    finally:
        _SCRIPTABLE, _EXPORTABLE, _NO_JIT = prev
