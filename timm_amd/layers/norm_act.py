"""Fused norm+act modules (reference `timm/layers/norm_act.py`).

`BatchNormAct2d` keeps norm/act as a single module so conv factories can treat
them as one unit (EfficientNet and friends depend on this structure).
Includes `convert_sync_batchnorm` (reference `:167`) and FrozenBatchNormAct2d.
"""
from typing import Optional, Type, Union

import functools

import torch
import torch.nn as nn
import torch.nn.functional as F

from .create_act import create_act_layer
from .norm import GroupNorm, GroupNorm1, LayerNorm, LayerNorm2d, RmsNorm, RmsNorm2d


def _create_act(act_layer, act_kwargs=None, inplace=False, apply_act=True):
    act_kwargs = act_kwargs or {}
    act_kwargs.setdefault('inplace', inplace)
    act = None
    if apply_act:
        act = create_act_layer(act_layer, **act_kwargs)
    return nn.Identity() if act is None else act


class BatchNormAct2d(nn.BatchNorm2d):
    """BatchNorm + Activation, keeping the same param naming as nn.BatchNorm2d
    so checkpoints interchange with the reference."""

    def __init__(
            self,
            num_features,
            eps=1e-5,
            momentum=0.1,
            affine=True,
            track_running_stats=True,
            apply_act=True,
            act_layer=nn.ReLU,
            act_kwargs=None,
            inplace=True,
            drop_layer=None,
            device=None,
            dtype=None,
    ):
        super().__init__(
            num_features, eps=eps, momentum=momentum, affine=affine,
            track_running_stats=track_running_stats, device=device, dtype=dtype)
        self.drop = drop_layer() if drop_layer is not None else nn.Identity()
        self.act = _create_act(act_layer, act_kwargs=act_kwargs, inplace=inplace, apply_act=apply_act)

    def forward(self, x):
        x = super().forward(x)
        x = self.drop(x)
        x = self.act(x)
        return x


class SyncBatchNormAct(nn.SyncBatchNorm):
    # Thanks to Selim Seferbekov (https://github.com/rwightman/pytorch-image-models/issues/1254)
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = super().forward(x)  # SyncBN doesn't work with torchscript anyways, so this is fine
        if hasattr(self, "drop"):
            x = self.drop(x)
        if hasattr(self, "act"):
            x = self.act(x)
        return x


def convert_sync_batchnorm(module, process_group=None):
    """Convert BatchNorm*/BatchNormAct* to SyncBatchNorm variants, preserving
    the act/drop of BatchNormAct2d (reference `norm_act.py:167`)."""
    module_output = module
    if isinstance(module, torch.nn.modules.batchnorm._BatchNorm):
        if isinstance(module, BatchNormAct2d):
            module_output = SyncBatchNormAct(
                module.num_features, module.eps, module.momentum,
                module.affine, module.track_running_stats, process_group=process_group)
            # set act and drop attr from the original module
            module_output.act = module.act
            module_output.drop = module.drop
        else:
            module_output = torch.nn.SyncBatchNorm(
                module.num_features, module.eps, module.momentum,
                module.affine, module.track_running_stats, process_group)
        if module.affine:
            with torch.no_grad():
                module_output.weight = module.weight
                module_output.bias = module.bias
        module_output.running_mean = module.running_mean
        module_output.running_var = module.running_var
        module_output.num_batches_tracked = module.num_batches_tracked
        if hasattr(module, 'qconfig'):
            module_output.qconfig = module.qconfig
    for name, child in module.named_children():
        module_output.add_module(name, convert_sync_batchnorm(child, process_group))
    del module
    return module_output


class FrozenBatchNormAct2d(torch.nn.Module):
    """BatchNormAct2d where the batch statistics and affine parameters are fixed
    (reference `norm_act.py:211`)."""

    def __init__(
            self,
            num_features: int,
            eps: float = 1e-5,
            apply_act=True,
            act_layer=nn.ReLU,
            act_kwargs=None,
            inplace=True,
            drop_layer=None,
    ):
        super().__init__()
        self.eps = eps
        self.register_buffer("weight", torch.ones(num_features))
        self.register_buffer("bias", torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.drop = drop_layer() if drop_layer is not None else nn.Identity()
        self.act = _create_act(act_layer, act_kwargs=act_kwargs, inplace=inplace, apply_act=apply_act)

    def _load_from_state_dict(
            self, state_dict, prefix, local_metadata, strict,
            missing_keys, unexpected_keys, error_msgs):
        num_batches_tracked_key = prefix + "num_batches_tracked"
        if num_batches_tracked_key in state_dict:
            del state_dict[num_batches_tracked_key]
        super()._load_from_state_dict(
            state_dict, prefix, local_metadata, strict, missing_keys, unexpected_keys, error_msgs)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        w = self.weight.reshape(1, -1, 1, 1)
        b = self.bias.reshape(1, -1, 1, 1)
        rv = self.running_var.reshape(1, -1, 1, 1)
        rm = self.running_mean.reshape(1, -1, 1, 1)
        scale = w * (rv + self.eps).rsqrt()
        bias = b - rm * scale
        x = x * scale + bias
        x = self.act(self.drop(x))
        return x

    def __repr__(self) -> str:
        return f"{self.__class__.__name__}({self.weight.shape[0]}, eps={self.eps}, act={self.act})"


def freeze_batch_norm_2d(module):
    """Recursively replace BatchNorm layers with FrozenBatchNormAct2d-style frozen variants."""
    res = module
    if isinstance(module, (BatchNormAct2d, SyncBatchNormAct)):
        res = FrozenBatchNormAct2d(module.num_features, module.eps)
        res.num_features = module.num_features
        res.weight.data = module.weight.data.clone().detach()
        res.bias.data = module.bias.data.clone().detach()
        res.running_mean.data = module.running_mean.data
        res.running_var.data = module.running_var.data
        res.drop = module.drop
        res.act = module.act
    elif isinstance(module, (torch.nn.modules.batchnorm.BatchNorm2d, torch.nn.modules.batchnorm.SyncBatchNorm)):
        res = FrozenBatchNormAct2d(module.num_features, module.eps, apply_act=False)
        res.num_features = module.num_features
        res.weight.data = module.weight.data.clone().detach()
        res.bias.data = module.bias.data.clone().detach()
        res.running_mean.data = module.running_mean.data
        res.running_var.data = module.running_var.data
    else:
        for name, child in module.named_children():
            new_child = freeze_batch_norm_2d(child)
            if new_child is not child:
                res.add_module(name, new_child)
    return res


def unfreeze_batch_norm_2d(module):
    res = module
    if isinstance(module, FrozenBatchNormAct2d):
        res = BatchNormAct2d(module.weight.shape[0], eps=module.eps)
        if module.weight is not None:
            res.weight.data = module.weight.data.clone().detach()
            res.bias.data = module.bias.data.clone().detach()
        res.running_mean.data = module.running_mean.data
        res.running_var.data = module.running_var.data
        res.drop = module.drop
        res.act = module.act
    else:
        for name, child in module.named_children():
            new_child = unfreeze_batch_norm_2d(child)
            if new_child is not child:
                res.add_module(name, new_child)
    return res


def _norm_act_map():
    # lazy: GroupNormAct/LayerNormAct* defined below in this module
    return dict(
        batchnorm=BatchNormAct2d,
        batchnorm2d=BatchNormAct2d,
        groupnorm=GroupNormAct,
        groupnorm1=functools.partial(GroupNormAct, num_groups=1),
        layernorm=LayerNormAct,
        layernorm2d=LayerNormAct2d,
        rmsnorm=RmsNormAct,
        rmsnorm2d=RmsNormAct2d,
        # evonorms fuse norm+act by construction
        evonormb0=_evo('EvoNorm2dB0'),
        evonormb1=_evo('EvoNorm2dB1'),
        evonormb2=_evo('EvoNorm2dB2'),
        evonorms0=_evo('EvoNorm2dS0'),
        evonorms0a=_evo('EvoNorm2dS0a'),
        evonorms1=_evo('EvoNorm2dS1'),
        evonorms1a=_evo('EvoNorm2dS1a'),
        evonorms2=_evo('EvoNorm2dS2'),
        evonorms2a=_evo('EvoNorm2dS2a'),
        frn=_evo('FilterResponseNormAct2d'),
        frntlu=_evo('FilterResponseNormTlu2d'),
    )


def _evo(name):
    from . import evo_norm, filter_response_norm
    return getattr(evo_norm, name, None) or getattr(filter_response_norm, name)
_NORM_ACT_TYPES = {BatchNormAct2d, SyncBatchNormAct, FrozenBatchNormAct2d}
# has act_layer arg to define act type
_NORM_ACT_REQUIRES_ARG = {BatchNormAct2d}


def get_norm_act_layer(norm_layer, act_layer=None):
    """Norm+act factory (reference `timm/layers/create_norm_act.py`)."""
    import functools
    import types
    if norm_layer is None:
        return None
    norm_act_kwargs = {}
    if isinstance(norm_layer, functools.partial):
        norm_act_kwargs.update(norm_layer.keywords)
        norm_layer = norm_layer.func
    if isinstance(norm_layer, str):
        if not norm_layer:
            return None
        layer_name = norm_layer.replace('_', '').lower().split('-')[0]
        norm_act_layer = _norm_act_map()[layer_name]
    elif norm_layer in _NORM_ACT_TYPES:
        norm_act_layer = norm_layer
    elif isinstance(norm_layer, types.FunctionType):
        norm_act_layer = norm_layer
    else:
        type_name = norm_layer.__name__.lower()
        if type_name.startswith('batchnorm'):
            norm_act_layer = BatchNormAct2d
        elif type_name.startswith('groupnormact') or type_name.startswith('layernormact') \
                or type_name.startswith('rmsnormact'):
            norm_act_layer = norm_layer  # already a norm+act type
        elif type_name == 'rmsnorm2d':
            norm_act_layer = RmsNormAct2d
        elif type_name == 'rmsnorm':
            norm_act_layer = RmsNormAct
        elif type_name == 'layernorm2d':
            norm_act_layer = LayerNormAct2d
        elif type_name == 'layernorm':
            norm_act_layer = LayerNormAct
        elif type_name.startswith('groupnorm'):
            norm_act_layer = _group_norm_act_factory
        elif type_name.startswith('evonorm') or type_name.startswith('filterresponsenorm'):
            norm_act_layer = norm_layer  # fused norm+act by construction
        else:
            raise AssertionError(f"No equivalent norm_act layer for {type_name}")

    base = norm_act_layer.func if isinstance(norm_act_layer, functools.partial) else norm_act_layer
    if base in _NORM_ACT_REQUIRES_ARG or base in (GroupNormAct, LayerNormAct, LayerNormAct2d, RmsNormAct, RmsNormAct2d):
        norm_act_kwargs.setdefault('act_layer', act_layer)
    if norm_act_kwargs:
        norm_act_layer = functools.partial(norm_act_layer, **norm_act_kwargs)
    return norm_act_layer


def _resolve_num_groups(num_channels, num_groups, group_size):
    """group_size (channels per group) takes precedence over num_groups."""
    if group_size:
        assert num_channels % group_size == 0
        return num_channels // group_size
    return num_groups


class GroupNormAct(GroupNorm):
    def __init__(
            self, num_channels, num_groups=32, eps=1e-5, affine=True,
            group_size=None,
            apply_act=True, act_layer=nn.ReLU, act_kwargs=None, inplace=True, drop_layer=None):
        super().__init__(
            num_channels,
            num_groups=_resolve_num_groups(num_channels, num_groups, group_size),
            eps=eps, affine=affine)
        self.drop = drop_layer() if drop_layer is not None else nn.Identity()
        self.act = _create_act(act_layer, act_kwargs=act_kwargs, inplace=inplace, apply_act=apply_act)

    def forward(self, x):
        x = F.group_norm(x, self.num_groups, self.weight, self.bias, self.eps)
        x = self.drop(x)
        x = self.act(x)
        return x


def _group_norm_act_factory(num_features, **kwargs):
    return GroupNormAct(num_features, **kwargs)


class LayerNormAct2d(LayerNorm2d):
    def __init__(
            self, num_channels, eps=1e-6, affine=True,
            apply_act=True, act_layer=nn.ReLU, act_kwargs=None, inplace=True, drop_layer=None):
        super().__init__(num_channels, eps=eps, affine=affine)
        self.drop = drop_layer() if drop_layer is not None else nn.Identity()
        self.act = _create_act(act_layer, act_kwargs=act_kwargs, inplace=inplace, apply_act=apply_act)

    def forward(self, x):
        x = super().forward(x)
        x = self.drop(x)
        x = self.act(x)
        return x


class LayerNormAct(nn.LayerNorm):
    """LayerNorm (last-dim) + act, for NLC tensors."""
    def __init__(
            self, normalization_shape, eps=1e-6, affine=True,
            apply_act=True, act_layer=nn.ReLU, act_kwargs=None, inplace=True, drop_layer=None):
        super().__init__(normalization_shape, eps=eps, elementwise_affine=affine)
        self.drop = drop_layer() if drop_layer is not None else nn.Identity()
        self.act = _create_act(act_layer, act_kwargs=act_kwargs, inplace=inplace, apply_act=apply_act)

    def forward(self, x):
        x = F.layer_norm(x, self.normalized_shape, self.weight, self.bias, self.eps)
        x = self.drop(x)
        x = self.act(x)
        return x


class RmsNormAct(RmsNorm):
    """RMSNorm + act for NLC tensors."""
    def __init__(
            self, num_channels, eps=1e-6, affine=True,
            apply_act=True, act_layer=nn.ReLU, act_kwargs=None, inplace=True, drop_layer=None):
        super().__init__(num_channels, eps=eps, affine=affine)
        self.drop = drop_layer() if drop_layer is not None else nn.Identity()
        self.act = _create_act(act_layer, act_kwargs=act_kwargs, inplace=inplace, apply_act=apply_act)

    def forward(self, x):
        x = super().forward(x)
        x = self.drop(x)
        x = self.act(x)
        return x


class RmsNormAct2d(RmsNorm2d):
    """RMSNorm (over C of NCHW) + act."""
    def __init__(
            self, num_channels, eps=1e-6, affine=True,
            apply_act=True, act_layer=nn.ReLU, act_kwargs=None, inplace=True, drop_layer=None):
        super().__init__(num_channels, eps=eps, affine=affine)
        self.drop = drop_layer() if drop_layer is not None else nn.Identity()
        self.act = _create_act(act_layer, act_kwargs=act_kwargs, inplace=inplace, apply_act=apply_act)

    def forward(self, x):
        x = super().forward(x)
        x = self.drop(x)
        x = self.act(x)
        return x


def create_norm_act_layer(layer_name, num_features, act_layer=None, apply_act=True, jit=False, **kwargs):
    """Instantiate a fused norm+act layer by name (reference
    `timm/layers/create_norm_act.py:create_norm_act_layer`)."""
    layer = get_norm_act_layer(layer_name, act_layer=act_layer)
    layer_instance = layer(num_features, apply_act=apply_act, **kwargs)
    if jit:
        import torch
        layer_instance = torch.jit.script(layer_instance)
    return layer_instance
