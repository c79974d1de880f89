"""MI355X-native op layer.

Every hot-path op in the reference framework (`timm`) is a stock PyTorch op —
see SURVEY.md §2.10 for the op-by-op map.  Here each of them dispatches to a
hand-written CDNA4 (gfx950) HIP kernel compiled into the in-tree extension
`timm_amd._C`:

 * layer_norm / rms_norm (+LayerScale fusion)          — fused one-pass, fp32 accum
 * attention (flash-style fused SDPA, bf16/fp16)       — MFMA 16x16x32, online softmax
 * bias_act (bias+GELU/SiLU epilogue after GEMM)
 * residual add (+ LayerScale + DropPath) epilogue
 * multi-tensor fused optimizer steps (AdamW/LAMB/SGD/lerp-EMA)

Dispatch rules:
 * CUDA/ROCm tensor + extension available  -> HIP kernel.
 * CUDA/ROCm tensor + extension missing    -> RuntimeError (loud, no silent
   eager fallback on a GPU box).
 * CPU tensor                              -> plain PyTorch reference path
   (keeps the full test-suite runnable without a GPU; this is also the
   numerics reference the GPU tests compare against).
"""
import os

import torch

_C = None
_C_ERR = None


def _load_extension():
    global _C, _C_ERR
    if _C is not None or _C_ERR is not None:
        return _C
    try:
        from timm_amd import _C as _ext  # built in-tree via setup.py build_ext --inplace
        _C = _ext
    except ImportError as e:
        _C_ERR = e
    return _C


def has_ext() -> bool:
    return _load_extension() is not None


def require_ext():
    ext = _load_extension()
    if ext is None:
        raise RuntimeError(
            "timm_amd HIP extension (timm_amd._C) is not built. On a GPU box this is a "
            "hard error - the native gfx950 kernels must run, not an eager fallback. "
            f"Build with `python setup.py build_ext --inplace`. Import error: {_C_ERR}")
    return ext


_ALLOW_EAGER_GPU = os.environ.get('TIMM_AMD_ALLOW_EAGER_GPU', '0') == '1'


def use_hip(x: torch.Tensor) -> bool:
    """True if op should take the HIP kernel path for tensor x."""
    if not x.is_cuda:
        return False
    if has_ext():
        return True
    if _ALLOW_EAGER_GPU:
        return False
    require_ext()  # raises


from .conv import depthwise_conv2d  # noqa: E402
from .layer_norm import layer_norm_act, rms_norm_act, layer_norm, rms_norm  # noqa: E402
from .attention import flash_attention, flash_attention_qkv, attention_available  # noqa: E402
from .elementwise import bias_act, residual_scale_add  # noqa: E402
from .fused_optim import fused_adamw_step, fused_lamb_step, fused_lerp_, fused_l2norm  # noqa: E402
from .loss import fused_cross_entropy  # noqa: E402
from .data import u8_normalize, masked_global_pool  # noqa: E402

__all__ = [
    'has_ext', 'require_ext', 'use_hip',
    'depthwise_conv2d',
    'layer_norm', 'layer_norm_act', 'rms_norm', 'rms_norm_act',
    'flash_attention', 'flash_attention_qkv', 'attention_available',
    'bias_act', 'residual_scale_add',
    'fused_adamw_step', 'fused_lamb_step', 'fused_lerp_', 'fused_l2norm',
    'fused_cross_entropy', 'u8_normalize', 'masked_global_pool',
]
