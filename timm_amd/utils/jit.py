"""torch.jit fuser selection (reference `timm/utils/jit.py:23`).

On ROCm the TE fuser is the only relevant scripted-graph fuser; nvfuser is a
CUDA-only backend and is always disabled.
"""
import torch


def _disable_nvfuser():
    try:
        torch._C._jit_set_nvfuser_enabled(False)
    except Exception:
        pass


def set_jit_legacy():
    """Legacy JIT executor with GPU op fusion."""
    assert hasattr(torch._C, '_jit_set_profiling_executor'), "Old JIT behavior doesn't exist!"
    torch._C._jit_set_profiling_executor(False)
    torch._C._jit_set_profiling_mode(False)
    torch._C._jit_override_can_fuse_on_gpu(True)


def set_jit_fuser(fuser):
    if fuser == 'te':
        torch._C._jit_set_profiling_executor(True)
        torch._C._jit_set_profiling_mode(True)
        torch._C._jit_override_can_fuse_on_cpu(False)
        torch._C._jit_override_can_fuse_on_gpu(True)
        torch._C._jit_set_texpr_fuser_enabled(True)
        _disable_nvfuser()
    elif fuser in ('old', 'legacy'):
        torch._C._jit_set_profiling_executor(False)
        torch._C._jit_set_profiling_mode(False)
        torch._C._jit_override_can_fuse_on_gpu(True)
        torch._C._jit_set_texpr_fuser_enabled(False)
        _disable_nvfuser()
    elif fuser in ('none', '', None):
        torch._C._jit_set_profiling_executor(True)
        torch._C._jit_set_profiling_mode(True)
        torch._C._jit_override_can_fuse_on_cpu(False)
        torch._C._jit_override_can_fuse_on_gpu(False)
        torch._C._jit_set_texpr_fuser_enabled(False)
        _disable_nvfuser()
    else:
        raise AssertionError(f'Invalid jit fuser ({fuser})')
