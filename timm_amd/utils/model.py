"""Model/state-dict utils (reference `timm/utils/model.py`)."""
import fnmatch
from copy import deepcopy
from typing import Callable, Optional, Sequence, Tuple, Union

import torch
from torch import nn as nn

from ..layers import BatchNormAct2d, FrozenBatchNormAct2d, freeze_batch_norm_2d, unfreeze_batch_norm_2d


def avg_sq_ch_mean(model, input, output):
    """Calculate average channel square mean of output activations."""
    return torch.mean(output.mean(axis=[0, 2, 3]) ** 2).item()


def avg_ch_var(model, input, output):
    return torch.mean(output.var(axis=[0, 2, 3])).item()


def avg_ch_var_residual(model, input, output):
    return torch.mean(output.var(axis=[0, 2, 3])).item()


def unwrap_model(model):
    """Unwrap DDP / EMA / compile wrappers to the bare model."""
    if isinstance(model, ModelEmaProxy):
        return unwrap_model(model.module)
    if hasattr(model, 'module') and not isinstance(model, torch.jit.ScriptModule):
        return unwrap_model(model.module)
    if hasattr(model, '_orig_mod'):
        return unwrap_model(model._orig_mod)
    return model


class ModelEmaProxy:
    # marker base to avoid circular import; real EMA classes subclass nn.Module with .module
    pass


def get_state_dict(model, unwrap_fn=unwrap_model):
    return unwrap_fn(model).state_dict()


class ActivationStatsHook:
    """Registers fwd hooks to compute per-module activation statistics
    (reference `utils/model.py:50`)."""

    def __init__(self, model, hook_fn_locs, hook_fns):
        self.model = model
        self.hook_fn_locs = hook_fn_locs
        self.hook_fns = hook_fns
        if len(hook_fn_locs) != len(hook_fns):
            raise ValueError("Please provide `hook_fns` for each `hook_fn_locs`, "
                             "their lengths are different.")
        self.stats = dict((hook_fn.__name__, []) for hook_fn in hook_fns)
        for hook_fn_loc, hook_fn in zip(hook_fn_locs, hook_fns):
            self.register_hook(hook_fn_loc, hook_fn)

    def _create_hook(self, hook_fn):
        def append_activation_stats(module, input, output):
            out = hook_fn(module, input, output)
            self.stats[hook_fn.__name__].append(out)
        return append_activation_stats

    def register_hook(self, hook_fn_loc, hook_fn):
        for name, module in self.model.named_modules():
            if not fnmatch.fnmatch(name, hook_fn_loc):
                continue
            module.register_forward_hook(self._create_hook(hook_fn))


def extract_spp_stats(model, hook_fn_locs, hook_fns, input_shape=[8, 3, 224, 224]):
    """Extract average square channel mean and variance of activations during
    forward pass to plot Signal Propagation Plots (SPP)."""
    x = torch.normal(0., 1., input_shape)
    hook = ActivationStatsHook(model, hook_fn_locs=hook_fn_locs, hook_fns=hook_fns)
    _ = model(x)
    return hook.stats


def _freeze_unfreeze(root_module, submodules=[], include_bn_running_stats=True, mode='freeze'):
    """Freeze or unfreeze parameters and/or BatchNorm buffers
    (reference `utils/model.py:181-232`)."""
    assert mode in ["freeze", "unfreeze"], '`mode` must be one of "freeze" or "unfreeze"'

    if isinstance(root_module, (
            torch.nn.modules.batchnorm.BatchNorm2d, torch.nn.modules.batchnorm.SyncBatchNorm,
            BatchNormAct2d)):
        # Raise assertion here because we can't convert it in place
        raise AssertionError(
            "You have provided a batch norm layer as the `root module`. Please use "
            "`timm_amd.utils.freeze_batch_norm_2d` or `timm_amd.utils.unfreeze_batch_norm_2d` instead.")

    if isinstance(submodules, str):
        submodules = [submodules]

    named_modules = submodules
    submodules = [root_module.get_submodule(m) for m in submodules]

    if not len(submodules):
        named_modules, submodules = list(zip(*root_module.named_children()))

    for n, m in zip(named_modules, submodules):
        # (Un)freeze parameters
        for p in m.parameters():
            p.requires_grad = False if mode == 'freeze' else True
        if include_bn_running_stats:
            # Helper to add submodule specified as a named_module
            def _add_submodule(module, name, submodule):
                split = name.rsplit('.', 1)
                if len(split) > 1:
                    module.get_submodule(split[0]).add_module(split[1], submodule)
                else:
                    module.add_module(name, submodule)

            if mode == 'freeze':
                res = freeze_batch_norm_2d(m)
                # It's possible that `m` is a type of BatchNorm in itself, in which case
                # `freeze_batch_norm_2d` won't convert it in place, but will return the converted result
                if isinstance(m, (
                        torch.nn.modules.batchnorm.BatchNorm2d,
                        torch.nn.modules.batchnorm.SyncBatchNorm, BatchNormAct2d)):
                    _add_submodule(root_module, n, res)
            else:
                res = unfreeze_batch_norm_2d(m)
                if isinstance(m, FrozenBatchNormAct2d):
                    _add_submodule(root_module, n, res)


def freeze(root_module, submodules=[], include_bn_running_stats=True):
    _freeze_unfreeze(root_module, submodules, include_bn_running_stats=include_bn_running_stats, mode="freeze")


def unfreeze(root_module, submodules=[], include_bn_running_stats=True):
    _freeze_unfreeze(root_module, submodules, include_bn_running_stats=include_bn_running_stats, mode="unfreeze")


def reparameterize_model(model: torch.nn.Module, inplace=False) -> torch.nn.Module:
    """Fuse reparameterizable branches (RepVGG/MobileOne style) for deploy
    (reference `utils/model.py:233`)."""
    if not inplace:
        model = deepcopy(model)

    def _fuse(m: torch.nn.Module):
        for child_name, child in m.named_children():
            if hasattr(child, 'fuse'):
                setattr(m, child_name, child.fuse())
            elif hasattr(child, "reparameterize"):
                child.reparameterize()
            elif hasattr(child, "switch_to_deploy"):
                child.switch_to_deploy()
            _fuse(child)

    _fuse(model)
    return model
