"""'Fast' (no-autocast-upcast) normalization API (reference
`timm/layers/fast_norm.py`, 259 LoC).

The reference uses this module to optionally skip AMP's fp32 upcast around
LN/GN/RMSNorm (and to call APEX fused kernels when installed). In this
framework the fused HIP norm kernels (`ops.layer_norm` / `ops.rms_norm`)
already run reduced-precision inputs with fp32 accumulators, so the "fast"
flag only controls whether the autocast-dtype downcast is applied before
the functional fallbacks here. The API surface (is_fast_norm/set_fast_norm
and the fast_* functions) is kept for parity with layers/models that
consult it.
"""
from typing import List, Optional

import torch
from torch.nn import functional as F

has_apex = False          # APEX is CUDA-only; never used on ROCm
has_apex_rmsnorm = False
has_torch_rms_norm = hasattr(F, 'rms_norm')

_USE_FAST_NORM = False


def get_autocast_dtype(device: str = 'cuda'):
    try:
        return torch.get_autocast_dtype(device)
    except (AttributeError, TypeError):
        if device == 'cpu':
            return torch.get_autocast_cpu_dtype()
        assert device == 'cuda'
        return torch.get_autocast_gpu_dtype()


def is_autocast_enabled(device: str = 'cuda'):
    try:
        return torch.is_autocast_enabled(device)
    except TypeError:
        if device == 'cpu':
            return torch.is_autocast_cpu_enabled()
        assert device == 'cuda'
        return torch.is_autocast_enabled()


def is_fast_norm():
    return _USE_FAST_NORM


def set_fast_norm(enable=True):
    global _USE_FAST_NORM
    _USE_FAST_NORM = enable


def _maybe_downcast(x, weight, bias, device_type):
    """When fast-norm is on under autocast, run at the autocast dtype
    instead of letting the norm op upcast to fp32."""
    if is_autocast_enabled(device_type):
        dt = get_autocast_dtype(device_type)
        x = x.to(dt)
        weight = weight.to(dt) if weight is not None else None
        bias = bias.to(dt) if bias is not None else None
    return x, weight, bias


def fast_group_norm(
        x: torch.Tensor,
        num_groups: int,
        weight: Optional[torch.Tensor] = None,
        bias: Optional[torch.Tensor] = None,
        eps: float = 1e-5,
) -> torch.Tensor:
    if torch.jit.is_scripting():
        return F.group_norm(x, num_groups, weight, bias, eps)
    x, weight, bias = _maybe_downcast(x, weight, bias, x.device.type)
    with torch.amp.autocast(device_type=x.device.type, enabled=False):
        return F.group_norm(x, num_groups, weight, bias, eps)


def fast_layer_norm(
        x: torch.Tensor,
        normalized_shape: List[int],
        weight: Optional[torch.Tensor] = None,
        bias: Optional[torch.Tensor] = None,
        eps: float = 1e-5,
) -> torch.Tensor:
    if torch.jit.is_scripting():
        return F.layer_norm(x, normalized_shape, weight, bias, eps)
    from .. import ops
    if x.is_cuda:
        return ops.layer_norm(x, normalized_shape, weight, bias, eps)
    x, weight, bias = _maybe_downcast(x, weight, bias, x.device.type)
    with torch.amp.autocast(device_type=x.device.type, enabled=False):
        return F.layer_norm(x, normalized_shape, weight, bias, eps)


def rms_norm(
        x: torch.Tensor,
        normalized_shape: List[int],
        weight: Optional[torch.Tensor] = None,
        eps: float = 1e-5,
):
    from .. import ops
    return ops.rms_norm(x, normalized_shape, weight, eps)


def fast_rms_norm(
        x: torch.Tensor,
        normalized_shape: List[int],
        weight: Optional[torch.Tensor] = None,
        eps: float = 1e-5,
) -> torch.Tensor:
    if torch.jit.is_scripting():
        return rms_norm(x, normalized_shape, weight, eps)
    x, weight, _ = _maybe_downcast(x, weight, None, x.device.type)
    return rms_norm(x, normalized_shape, weight, eps)


def rms_norm2d(
        x: torch.Tensor,
        normalized_shape: List[int],
        weight: Optional[torch.Tensor] = None,
        eps: float = 1e-5,
):
    assert len(normalized_shape) == 1
    x = x.permute(0, 2, 3, 1)
    x = rms_norm(x, normalized_shape, weight, eps)
    return x.permute(0, 3, 1, 2)


def fast_rms_norm2d(
        x: torch.Tensor,
        normalized_shape: List[int],
        weight: Optional[torch.Tensor] = None,
        eps: float = 1e-5,
) -> torch.Tensor:
    x, weight, _ = _maybe_downcast(x, weight, None, x.device.type)
    return rms_norm2d(x, normalized_shape, weight, eps)


def simple_norm(
        x: torch.Tensor,
        normalized_shape: List[int],
        weight: Optional[torch.Tensor] = None,
        eps: float = 1e-5,
):
    # RMSNorm without mean-centering or bias; same compute path here
    return rms_norm(x, normalized_shape, weight, eps)


def fast_simple_norm(
        x: torch.Tensor,
        normalized_shape: List[int],
        weight: Optional[torch.Tensor] = None,
        eps: float = 1e-5,
) -> torch.Tensor:
    x, weight, _ = _maybe_downcast(x, weight, None, x.device.type)
    return simple_norm(x, normalized_shape, weight, eps)
