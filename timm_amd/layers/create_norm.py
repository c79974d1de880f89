"""Norm-layer factory (reference `timm/layers/create_norm.py`): name or
callable -> norm constructor, preserving partial-bound kwargs."""
import functools
import types

import torch.nn as nn

from .norm import (
    GroupNorm, GroupNorm1, LayerNorm, LayerNorm2d, RmsNorm, RmsNorm2d, SimpleNorm, SimpleNorm2d,
)

_NORM_MAP = dict(
    batchnorm=nn.BatchNorm2d,
    batchnorm2d=nn.BatchNorm2d,
    batchnorm1d=nn.BatchNorm1d,
    groupnorm=GroupNorm,
    groupnorm1=GroupNorm1,
    layernorm=LayerNorm,
    layernorm2d=LayerNorm2d,
    rmsnorm=RmsNorm,
    rmsnorm2d=RmsNorm2d,
    simplenorm=SimpleNorm,
    simplenorm2d=SimpleNorm2d,
)
_NORM_TYPES = set(_NORM_MAP.values())


def create_norm_layer(layer_name, num_features, **kwargs):
    return get_norm_layer(layer_name)(num_features, **kwargs)


def get_norm_layer(norm_layer):
    if norm_layer is None:
        return None
    assert isinstance(norm_layer, (type, str, types.FunctionType, functools.partial))

    bound_kwargs = {}
    if isinstance(norm_layer, functools.partial):
        # unwrap so any bound kwargs survive the name lookup and re-bind below
        bound_kwargs.update(norm_layer.keywords)
        norm_layer = norm_layer.func

    if isinstance(norm_layer, str):
        if not norm_layer:
            return None
        norm_layer = _NORM_MAP[norm_layer.replace('_', '').lower()]

    if bound_kwargs:
        norm_layer = functools.partial(norm_layer, **bound_kwargs)
    return norm_layer
