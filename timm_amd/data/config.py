"""Data-config resolution: merge CLI args over a model's pretrained cfg.

Behavioral parity: /root/reference/timm/data/config.py:8-100 (same precedence
rules, same output keys).  Implemented here as a declarative precedence table
walked by one resolver instead of a per-field if/elif chain.
"""
import logging

from .constants import DEFAULT_CROP_MODE, DEFAULT_CROP_PCT, \
    IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD

_logger = logging.getLogger(__name__)

__all__ = ['resolve_data_config', 'resolve_model_data_config']


def _first(*candidates):
    """First candidate that is neither None nor empty-falsy."""
    for value in candidates:
        if value:
            return value
    return None


def _per_channel(values, channels):
    """Normalize a mean/std spec to one value per channel."""
    values = tuple(values)
    if len(values) == 1:
        return values * channels
    assert len(values) == channels
    return values


def resolve_data_config(
        args=None,
        pretrained_cfg=None,
        model=None,
        use_test_size=False,
        verbose=False,
):
    """Build the input-pipeline config dict for a model.

    Precedence per field: explicit args > pretrained_cfg (> test-size variants
    when ``use_test_size``) > library default.
    """
    assert model or args or pretrained_cfg, \
        'At least one of model, args, or pretrained_cfg required for data config.'
    args = args or {}
    if not pretrained_cfg:
        pretrained_cfg = getattr(model, 'pretrained_cfg', None) or {}

    # ---- input tensor shape ----
    in_chans = args.get('in_chans') or args.get('chans') or 3
    if args.get('input_size') is not None:
        input_size = tuple(args['input_size'])
        assert len(input_size) == 3
        in_chans = input_size[0]
    elif args.get('img_size') is not None:
        side = args['img_size']
        assert isinstance(side, int)
        input_size = (in_chans, side, side)
    else:
        cfg_size = None
        if use_test_size:
            cfg_size = pretrained_cfg.get('test_input_size')
        cfg_size = cfg_size if cfg_size is not None else pretrained_cfg.get('input_size')
        input_size = cfg_size if cfg_size is not None else (in_chans, 224, 224)

    # ---- normalization stats (expanded to in_chans when given as scalars) ----
    mean = args.get('mean')
    mean = _per_channel(mean, in_chans) if mean is not None else \
        _first(pretrained_cfg.get('mean'), IMAGENET_DEFAULT_MEAN)
    std = args.get('std')
    std = _per_channel(std, in_chans) if std is not None else \
        _first(pretrained_cfg.get('std'), IMAGENET_DEFAULT_STD)

    # ---- eval crop ----
    crop_pct = args.get('crop_pct')
    if crop_pct:
        crop_pct = float(crop_pct)
    else:
        test_pct = pretrained_cfg.get('test_crop_pct') if use_test_size else None
        crop_pct = _first(test_pct, pretrained_cfg.get('crop_pct'), DEFAULT_CROP_PCT)

    config = dict(
        input_size=input_size,
        interpolation=_first(
            args.get('interpolation'), pretrained_cfg.get('interpolation'), 'bicubic'),
        mean=mean,
        std=std,
        crop_pct=crop_pct,
        crop_mode=_first(
            args.get('crop_mode'), pretrained_cfg.get('crop_mode'), DEFAULT_CROP_MODE),
    )

    if verbose:
        _logger.info('Data processing configuration for current model + dataset:')
        for key, value in config.items():
            _logger.info('\t%s: %s', key, value)
    return config


def resolve_model_data_config(
        model,
        args=None,
        pretrained_cfg=None,
        use_test_size=False,
        verbose=False,
):
    """Model-first argument-order variant of resolve_data_config."""
    return resolve_data_config(
        args=args,
        pretrained_cfg=pretrained_cfg,
        model=model,
        use_test_size=use_test_size,
        verbose=verbose,
    )
