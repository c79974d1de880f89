"""Shared typing aliases + compiler-disable shim (reference `timm/layers/typing.py`)."""
from contextlib import nullcontext
from functools import wraps
from typing import Callable, ContextManager, Optional, Tuple, Type, TypeVar, Union, overload

import torch

__all__ = ['LayerType', 'PadType', 'nullwrap', 'disable_compiler']

LayerType = Union[str, Callable, Type[torch.nn.Module]]
PadType = Union[str, int, Tuple[int, int]]

F = TypeVar('F', bound=Callable[..., object])


@overload
def nullwrap(fn: F) -> F: ...


@overload
def nullwrap(fn: None = ...) -> ContextManager: ...


def nullwrap(fn: Optional[F] = None):
    """No-op usable both as decorator and context manager."""
    if fn is None:
        return nullcontext()

    @wraps(fn)
    def wrapper(*args, **kwargs):
        return fn(*args, **kwargs)
    return wrapper


disable_compiler = getattr(getattr(torch, 'compiler', None), 'disable', None) or nullwrap
