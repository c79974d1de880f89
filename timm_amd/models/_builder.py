"""Shared model-build funnel (reference `timm/models/_builder.py:384-503`).

Every architecture entrypoint calls `build_model_with_cfg`: it resolves the
pretrained cfg (registry lookup + overlay), seeds model kwargs from it,
instantiates the class, optionally loads+adapts pretrained weights (input
conv channel conversion, classifier drop/offset), and wraps for
features_only extraction.
"""
import dataclasses
import logging
import os
from copy import deepcopy
from typing import Any, Callable, Dict, Optional, Tuple

import torch
from torch import nn

from ._features import FeatureDictNet, FeatureGetterNet, FeatureHookNet, FeatureListNet
from ._helpers import load_state_dict
from ._manipulate import adapt_input_conv
from ._pretrained import PretrainedCfg
from ._registry import get_pretrained_cfg

_logger = logging.getLogger(__name__)

# process-wide download/load knobs
_DOWNLOAD_PROGRESS = False
_CHECK_HASH = False
_USE_OLD_CACHE = int(os.environ.get('TIMM_USE_OLD_CACHE', 0)) > 0

__all__ = ['set_pretrained_download_progress', 'set_pretrained_check_hash', 'load_custom_pretrained',
           'load_pretrained', 'pretrained_cfg_for_features', 'resolve_pretrained_cfg', 'build_model_with_cfg']


def set_pretrained_download_progress(enable=True):
    global _DOWNLOAD_PROGRESS
    _DOWNLOAD_PROGRESS = enable


def set_pretrained_check_hash(enable=True):
    global _CHECK_HASH
    _CHECK_HASH = enable


def _resolve_pretrained_source(pretrained_cfg: Dict[str, Any]) -> Tuple[str, Any]:
    """Pick (load_from, location). Priority: explicit hf-hub source, then
    in-cfg state_dict > file > url > hf_hub_id."""
    cfg_source = pretrained_cfg.get('source', '')
    in_cfg_sd = pretrained_cfg.get('state_dict', None)
    url = pretrained_cfg.get('url', None)
    file = pretrained_cfg.get('file', None)
    hf_hub_id = pretrained_cfg.get('hf_hub_id', None)

    load_from, location = '', ''
    if cfg_source == 'hf-hub' and hf_hub_id:
        load_from, location = 'hf-hub', hf_hub_id
    elif in_cfg_sd:
        assert isinstance(in_cfg_sd, dict)
        load_from, location = 'state_dict', in_cfg_sd
    elif file:
        load_from, location = 'file', file
    elif url:
        load_from, location = 'url', url
    elif hf_hub_id:
        load_from, location = 'hf-hub', hf_hub_id

    if load_from == 'hf-hub' and pretrained_cfg.get('hf_hub_filename', None):
        location = (location, pretrained_cfg['hf_hub_filename'])
    return load_from, location


def load_custom_pretrained(
        model: nn.Module,
        pretrained_cfg: Optional[Dict] = None,
        load_fn: Optional[Callable] = None,
        cache_dir: Optional[str] = None,
):
    """Load weights through a model-provided `load_pretrained` fn (npz-style
    custom checkpoints)."""
    pretrained_cfg = pretrained_cfg or getattr(model, 'pretrained_cfg', None)
    if not pretrained_cfg:
        _logger.warning('Invalid pretrained config, cannot load weights.')
        return

    load_from, location = _resolve_pretrained_source(pretrained_cfg)
    if not load_from:
        _logger.warning('No pretrained weights exist for this model. Using random initialization.')
        return
    if load_from == 'hf-hub':
        _logger.warning('Hugging Face hub not currently supported for custom load pretrained models.')
    elif load_from == 'url':
        from ._hub import download_cached_file
        location = download_cached_file(location, check_hash=_CHECK_HASH, cache_dir=cache_dir)

    if load_fn is not None:
        load_fn(model, location)
    elif hasattr(model, 'load_pretrained'):
        model.load_pretrained(location)
    else:
        _logger.warning('Valid function to load pretrained weights is not available, using random initialization.')


def _fetch_state_dict(model, pretrained_cfg, cache_dir):
    """Materialize the pretrained state dict (or None if the model's own
    custom loader consumed it)."""
    load_from, location = _resolve_pretrained_source(pretrained_cfg)
    if load_from == 'state_dict':
        _logger.info('Loading pretrained weights from state dict')
        return location
    if load_from == 'file':
        _logger.info(f'Loading pretrained weights from file ({location})')
        if pretrained_cfg.get('custom_load', False):
            model.load_pretrained(location)
            return None
        return load_state_dict(location)
    if load_from == 'url':
        _logger.info(f'Loading pretrained weights from url ({location})')
        if pretrained_cfg.get('custom_load', False):
            from ._hub import download_cached_file
            location = download_cached_file(
                location, progress=_DOWNLOAD_PROGRESS, check_hash=_CHECK_HASH, cache_dir=cache_dir)
            model.load_pretrained(location)
            return None
        try:
            return torch.hub.load_state_dict_from_url(
                location, map_location='cpu', progress=_DOWNLOAD_PROGRESS,
                check_hash=_CHECK_HASH, weights_only=True)
        except TypeError:  # older torch without weights_only
            return torch.hub.load_state_dict_from_url(
                location, map_location='cpu', progress=_DOWNLOAD_PROGRESS, check_hash=_CHECK_HASH)
    if load_from == 'hf-hub':
        _logger.info(f'Loading pretrained weights from Hugging Face hub ({location})')
        from ._hub import load_state_dict_from_hf
        if isinstance(location, (list, tuple)):
            return load_state_dict_from_hf(*location, cache_dir=cache_dir)
        return load_state_dict_from_hf(location, cache_dir=cache_dir, weights_only=True)
    model_name = pretrained_cfg.get('architecture', 'this model')
    raise RuntimeError(f'No pretrained weights exist for {model_name}. Use `pretrained=False` for random init.')


def load_pretrained(
        model: nn.Module,
        pretrained_cfg: Optional[Dict] = None,
        num_classes: int = 1000,
        in_chans: int = 3,
        filter_fn: Optional[Callable] = None,
        strict: bool = True,
        cache_dir: Optional[str] = None,
):
    """Load + adapt a pretrained checkpoint (reference `_builder.py:152-288`):
    stem conv converted for in_chans != 3, classifier dropped on class-count
    mismatch or sliced on label_offset."""
    pretrained_cfg = pretrained_cfg or getattr(model, 'pretrained_cfg', None)
    if not pretrained_cfg:
        raise RuntimeError('Invalid pretrained config, cannot load weights. Use `pretrained=False` for random init.')

    state_dict = _fetch_state_dict(model, pretrained_cfg, cache_dir)
    if state_dict is None:
        return  # custom loader already applied the weights

    if filter_fn is not None:
        try:
            state_dict = filter_fn(state_dict, model)
        except TypeError:
            state_dict = filter_fn(state_dict)  # single-arg legacy filter

    input_convs = pretrained_cfg.get('first_conv', None)
    if input_convs is not None and in_chans != 3:
        if isinstance(input_convs, str):
            input_convs = (input_convs,)
        for conv_name in input_convs:
            weight_name = conv_name + '.weight'
            try:
                state_dict[weight_name] = adapt_input_conv(in_chans, state_dict[weight_name])
                _logger.info(
                    f'Converted input conv {conv_name} pretrained weights from 3 to {in_chans} channel(s)')
            except NotImplementedError:
                del state_dict[weight_name]
                strict = False
                _logger.warning(
                    f'Unable to convert pretrained {conv_name} weights, using random init for this layer.')

    classifiers = pretrained_cfg.get('classifier', None)
    label_offset = pretrained_cfg.get('label_offset', 0)
    if classifiers is not None:
        if isinstance(classifiers, str):
            classifiers = (classifiers,)
        if num_classes != pretrained_cfg['num_classes']:
            # class count differs: drop the head weights entirely
            for name in classifiers:
                state_dict.pop(name + '.weight', None)
                state_dict.pop(name + '.bias', None)
            strict = False
        elif label_offset > 0:
            # checkpoints with leading background class(es): slice them off
            for name in classifiers:
                state_dict[name + '.weight'] = state_dict[name + '.weight'][label_offset:]
                state_dict[name + '.bias'] = state_dict[name + '.bias'][label_offset:]

    load_result = model.load_state_dict(state_dict, strict=strict)
    if load_result.missing_keys:
        _logger.info(
            f'Missing keys ({", ".join(load_result.missing_keys)}) discovered while loading pretrained weights.'
            f' This is expected if model is being adapted.')
    if load_result.unexpected_keys:
        _logger.warning(
            f'Unexpected keys ({", ".join(load_result.unexpected_keys)}) found while loading pretrained weights.'
            f' This may be expected if model is being adapted.')


def pretrained_cfg_for_features(pretrained_cfg):
    pretrained_cfg = deepcopy(pretrained_cfg)
    for field in ('num_classes', 'classifier', 'global_pool'):  # head-only fields
        pretrained_cfg.pop(field, None)
    return pretrained_cfg


def _filter_kwargs(kwargs, names):
    if not kwargs or not names:
        return
    for n in names:
        kwargs.pop(n, None)


def _update_default_model_kwargs(pretrained_cfg, kwargs, kwargs_filter):
    """Seed model kwargs from the pretrained cfg where not explicitly given.

    pretrained_cfg carries input_size=(C, H, W); models take img_size and
    in_chans separately, so those are split out here.
    """
    default_kwarg_names = ('num_classes', 'global_pool', 'in_chans')
    if pretrained_cfg.get('fixed_input_size', False):
        default_kwarg_names += ('img_size',)  # fixed-size models need it at build time

    for name in default_kwarg_names:
        if name == 'img_size':
            input_size = pretrained_cfg.get('input_size', None)
            if input_size is not None:
                assert len(input_size) == 3
                kwargs.setdefault(name, input_size[-2:])
        elif name == 'in_chans':
            input_size = pretrained_cfg.get('input_size', None)
            if input_size is not None:
                assert len(input_size) == 3
                kwargs.setdefault(name, input_size[0])
        elif name == 'num_classes':
            default_val = pretrained_cfg.get(name, None)
            if default_val is not None and default_val >= 0:  # negative = don't forward
                kwargs.setdefault(name, pretrained_cfg[name])
        else:
            default_val = pretrained_cfg.get(name, None)
            if default_val is not None:
                kwargs.setdefault(name, pretrained_cfg[name])

    _filter_kwargs(kwargs, names=kwargs_filter)


def resolve_pretrained_cfg(
        variant: str,
        pretrained_cfg=None,
        pretrained_cfg_overlay=None,
) -> PretrainedCfg:
    """Resolve a PretrainedCfg from an explicit dict / tag string / registry
    lookup, then apply the overlay."""
    model_with_tag = variant
    pretrained_tag = None
    if pretrained_cfg:
        if isinstance(pretrained_cfg, dict):
            pretrained_cfg = PretrainedCfg(**pretrained_cfg)  # validate
        elif isinstance(pretrained_cfg, str):
            pretrained_tag = pretrained_cfg
            pretrained_cfg = None

    if not pretrained_cfg:
        if pretrained_tag:
            model_with_tag = '.'.join([variant, pretrained_tag])
        pretrained_cfg = get_pretrained_cfg(model_with_tag)

    if not pretrained_cfg:
        _logger.warning(
            f'No pretrained configuration specified for {model_with_tag} model. Using a default.'
            f' Please add a config to the model pretrained_cfg registry or pass explicitly.')
        pretrained_cfg = PretrainedCfg()

    pretrained_cfg_overlay = pretrained_cfg_overlay or {}
    if not pretrained_cfg.architecture:
        pretrained_cfg_overlay.setdefault('architecture', variant)
    return dataclasses.replace(pretrained_cfg, **pretrained_cfg_overlay)


_FEATURE_CLS_MAP = {
    'list': FeatureListNet,
    'dict': FeatureDictNet,
    'getter': FeatureGetterNet,
}


def _resolve_feature_cls(feature_cfg):
    """(feature class, is_getter) from feature_cfg; default list-style."""
    if 'feature_cls' not in feature_cfg:
        return FeatureListNet, False
    feature_cls = feature_cfg.pop('feature_cls')
    if not isinstance(feature_cls, str):
        return feature_cls, False
    feature_cls = feature_cls.lower()
    if feature_cls not in ('dict', 'list', 'hook'):
        feature_cfg.pop('flatten_sequential', None)  # only the rebuilders accept it
    if 'hook' in feature_cls:
        return FeatureHookNet, False
    if feature_cls == 'fx':
        from ._features_fx import FeatureGraphNet
        return FeatureGraphNet, False
    assert feature_cls in _FEATURE_CLS_MAP, f'Unknown feature class {feature_cls}'
    return _FEATURE_CLS_MAP[feature_cls], feature_cls == 'getter'


def build_model_with_cfg(
        model_cls: Callable,
        variant: str,
        pretrained: bool,
        pretrained_cfg: Optional[Dict] = None,
        pretrained_cfg_overlay: Optional[Dict] = None,
        model_cfg: Optional[Any] = None,
        feature_cfg: Optional[Dict] = None,
        pretrained_strict: bool = True,
        pretrained_filter_fn: Optional[Callable] = None,
        cache_dir: Optional[str] = None,
        kwargs_filter: Optional[Tuple[str, ...]] = None,
        **kwargs,
):
    """The build funnel (reference `_builder.py:384-503`); see module docs."""
    pruned = kwargs.pop('pruned', False)
    features = False
    feature_cfg = feature_cfg or {}

    pretrained_cfg = resolve_pretrained_cfg(
        variant,
        pretrained_cfg=pretrained_cfg,
        pretrained_cfg_overlay=pretrained_cfg_overlay,
    ).to_dict()

    _update_default_model_kwargs(pretrained_cfg, kwargs, kwargs_filter)

    if kwargs.pop('features_only', False):
        features = True
        feature_cfg.setdefault('out_indices', (0, 1, 2, 3, 4))
        if 'out_indices' in kwargs:
            feature_cfg['out_indices'] = kwargs.pop('out_indices')
        if 'feature_cls' in kwargs:
            feature_cfg['feature_cls'] = kwargs.pop('feature_cls')

    model = model_cls(**kwargs) if model_cfg is None else model_cls(cfg=model_cfg, **kwargs)
    model.pretrained_cfg = pretrained_cfg
    model.default_cfg = model.pretrained_cfg  # backwards-compat alias

    if pruned:
        from ._prune import adapt_model_from_file
        model = adapt_model_from_file(model, variant)

    # class count the checkpoint head is compared against (0 for feature use)
    num_classes_pretrained = 0 if features else getattr(model, 'num_classes', kwargs.get('num_classes', 1000))
    if pretrained:
        load_pretrained(
            model,
            pretrained_cfg=pretrained_cfg,
            num_classes=num_classes_pretrained,
            in_chans=kwargs.get('in_chans', 3),
            filter_fn=pretrained_filter_fn,
            strict=pretrained_strict,
            cache_dir=cache_dir,
        )

    if features:
        feature_cls, use_getter = _resolve_feature_cls(feature_cfg)
        output_fmt = getattr(model, 'output_fmt', None)
        if output_fmt is not None and not use_getter:  # getter resolves fmt itself
            feature_cfg.setdefault('output_fmt', output_fmt)
        model = feature_cls(model, **feature_cfg)
        model.pretrained_cfg = pretrained_cfg_for_features(pretrained_cfg)
        model.default_cfg = model.pretrained_cfg

    return model
