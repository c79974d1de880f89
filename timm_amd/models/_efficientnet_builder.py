"""EfficientNet-family trunk builder + block-spec string DSL.

Behavioral parity: /root/reference/timm/models/_efficientnet_builder.py
(the "ir_r2_k3_s2_e6_c24_se0.25" grammar, depth scaling, space2depth
regions, output-stride dilation, feature-info bookkeeping, TF weight init).

Redesign: block decoding and construction are table-driven — one
``_BLOCK_KWARGS`` decoder map and one ``_BLOCK_FACTORY`` constructor map per
block type — rather than parallel if/elif ladders.
"""
import logging
import math
import re
from copy import deepcopy
from functools import partial
from typing import Any, Dict, List

import torch.nn as nn

from ..layers import get_act_layer, get_norm_act_layer, make_divisible
from ._efficientnet_blocks import (
    SqueezeExcite, ConvBnAct, DepthwiseSeparableConv, InvertedResidual, CondConvResidual,
    EdgeResidual, UniversalInvertedResidual, MobileAttention,
)
from ._manipulate import named_modules

__all__ = [
    'EfficientNetBuilder', 'BlockArgs', 'decode_arch_def', 'efficientnet_init_weights',
    'resolve_bn_args', 'resolve_act_layer', 'round_channels', 'BN_MOMENTUM_TF_DEFAULT',
    'BN_EPS_TF_DEFAULT',
]

_logger = logging.getLogger(__name__)

_DEBUG_BUILDER = False

# TF-style BN defaults (PT momentum = 1 - TF decay)
BN_MOMENTUM_TF_DEFAULT = 1 - 0.99
BN_EPS_TF_DEFAULT = 1e-3

BlockArgs = List[List[Dict[str, Any]]]


def get_bn_args_tf():
    return dict(momentum=BN_MOMENTUM_TF_DEFAULT, eps=BN_EPS_TF_DEFAULT)


def resolve_bn_args(kwargs):
    out = {}
    momentum = kwargs.pop('bn_momentum', None)
    if momentum is not None:
        out['momentum'] = momentum
    eps = kwargs.pop('bn_eps', None)
    if eps is not None:
        out['eps'] = eps
    return out


def resolve_act_layer(kwargs, default='relu'):
    return get_act_layer(kwargs.pop('act_layer', default))


def round_channels(channels, multiplier=1.0, divisor=8, channel_min=None, round_limit=0.9):
    """Width-multiplier channel rounding (divisor-aligned)."""
    if not multiplier:
        return channels
    return make_divisible(channels * multiplier, divisor, channel_min, round_limit=round_limit)


def _log_info_if(msg, condition):
    if condition:
        _logger.info(msg)


# ---------------------------------------------------------------------------
# block-spec DSL: "<type>_<key><val>_..." tokens
# ---------------------------------------------------------------------------

# activation shorthand used by the 'n' option
_ACT_SHORTHAND = dict(re='relu', r6='relu6', hs='hard_swish', sw='swish', mi='mish')


def _parse_ksize(v):
    """Kernel spec: '3' -> 3, '3.5.7' -> [3, 5, 7] (MixedConv)."""
    return int(v) if v.isdigit() else [int(k) for k in v.split('.')]


def _tokenize(ops):
    """Parse option tokens -> (options dict, skip override)."""
    options, skip = {}, None
    for op in ops:
        if op == 'noskip':
            skip = False
        elif op == 'skip':
            skip = True
        elif op.startswith('n'):
            act_name = _ACT_SHORTHAND.get(op[1:])
            if act_name is not None:
                options['n'] = get_act_layer(act_name)
        else:
            parts = re.split(r'(\d.*)', op)
            if len(parts) >= 2:
                options[parts[0]] = parts[1]
    return options, skip


def _kwargs_ir(o, skip, start_k, end_k):
    out = dict(
        dw_kernel_size=_parse_ksize(o['k']),
        exp_kernel_size=start_k,
        pw_kernel_size=end_k,
        exp_ratio=float(o['e']),
        se_ratio=float(o.get('se', 0.)),
        noskip=skip is False,
        s2d=int(o.get('d', 0)) > 0,
    )
    if 'cc' in o:
        out['num_experts'] = int(o['cc'])
    return out


def _kwargs_ds(o, skip, start_k, end_k, block_type='ds'):
    return dict(
        dw_kernel_size=_parse_ksize(o['k']),
        pw_kernel_size=end_k,
        se_ratio=float(o.get('se', 0.)),
        pw_act=block_type == 'dsa',
        noskip=block_type == 'dsa' or skip is False,
        s2d=int(o.get('d', 0)) > 0,
    )


def _kwargs_er(o, skip, start_k, end_k):
    return dict(
        exp_kernel_size=_parse_ksize(o['k']),
        pw_kernel_size=end_k,
        exp_ratio=float(o['e']),
        force_in_chs=int(o.get('fc', 0)),
        se_ratio=float(o.get('se', 0.)),
        noskip=skip is False,
    )


def _kwargs_cn(o, skip, start_k, end_k):
    return dict(kernel_size=int(o['k']), skip=skip is True)


def _kwargs_uir(o, skip, start_k, end_k):
    # uir reinterprets the a/p kernels as start/end dw kernels (0 = absent)
    return dict(
        dw_kernel_size_start=_parse_ksize(o['a']) if 'a' in o else 0,
        dw_kernel_size_mid=_parse_ksize(o['k']),
        dw_kernel_size_end=_parse_ksize(o['p']) if 'p' in o else 0,
        exp_ratio=float(o['e']),
        se_ratio=float(o.get('se', 0.)),
        noskip=skip is False,
    )


def _kwargs_attn(o, skip, start_k, end_k):
    kv_dim = int(o['d'])
    return dict(
        dw_kernel_size=_parse_ksize(o['k']),
        num_heads=int(o['h']),
        key_dim=kv_dim,
        value_dim=kv_dim,
        kv_stride=int(o.get('v', 1)),
        noskip=skip is False,
    )


_BLOCK_KWARGS = {
    'ir': _kwargs_ir,
    'ds': partial(_kwargs_ds, block_type='ds'),
    'dsa': partial(_kwargs_ds, block_type='dsa'),
    'er': _kwargs_er,
    'cn': _kwargs_cn,
    'uir': _kwargs_uir,
    'mha': _kwargs_attn,
    'mqa': _kwargs_attn,
}


def _decode_block_str(block_str):
    """One spec string -> (block kwargs, repeat count).

    Grammar: leading token = block type (ir/ds/dsa/er/cn/uir/mha/mqa), then
    r repeats · k kernel · s stride · e expansion · c out-chans · se ratio ·
    n act ('re','r6','hs','sw','mi') · a/p start/end kernels · d (s2d or
    attn kv-dim) · h heads · v kv-stride · cc experts · gs group size ·
    skip/noskip.
    """
    assert isinstance(block_str, str)
    block_type, *ops = block_str.split('_')
    assert block_type in _BLOCK_KWARGS, f'Unknown block type ({block_type})'
    options, skip = _tokenize(ops)

    start_k = _parse_ksize(options['a']) if 'a' in options else 1
    end_k = _parse_ksize(options['p']) if 'p' in options else 1

    args = dict(
        block_type=block_type,
        out_chs=int(options['c']),
        stride=int(options['s']),
        act_layer=options.get('n'),  # None -> model default at build time
    )
    args.update(_BLOCK_KWARGS[block_type](options, skip, start_k, end_k))
    if 'gs' in options:
        args['group_size'] = int(options['gs'])
    return args, int(options['r'])


def _scale_stage_depth(stack_args, repeats, depth_multiplier=1.0, depth_trunc='ceil'):
    """Scale a stage's total repeat count, distributing over its block defs.

    Allocation runs in reverse so the first (often unique) block of a stage
    is the last to gain repeats — matches EfficientNet compound scaling.
    """
    total = sum(repeats)
    if depth_trunc == 'round':
        # rounding keeps small stages proportionally small for longer
        total_scaled = max(1, round(total * depth_multiplier))
    else:
        # EfficientNet default: ceil, so any multiplier > 1 deepens all stages
        total_scaled = int(math.ceil(total * depth_multiplier))

    scaled = []
    remaining, remaining_scaled = total, total_scaled
    for r in reversed(repeats):
        take = max(1, round(r / remaining * remaining_scaled))
        scaled.append(take)
        remaining -= r
        remaining_scaled -= take
    scaled.reverse()

    out = []
    for args, rep in zip(stack_args, scaled):
        out.extend(deepcopy(args) for _ in range(rep))
    return out


def decode_arch_def(
        arch_def,
        depth_multiplier=1.0,
        depth_trunc='ceil',
        experts_multiplier=1,
        fix_first_last=False,
        group_size=None,
):
    """Expand per-stage spec-string lists into per-block kwargs lists."""
    if not isinstance(depth_multiplier, tuple):
        depth_multiplier = (depth_multiplier,) * len(arch_def)
    else:
        assert len(depth_multiplier) == len(arch_def)

    stages = []
    for stage_idx, (specs, mult) in enumerate(zip(arch_def, depth_multiplier)):
        assert isinstance(specs, list)
        stage_args, repeats = [], []
        for spec in specs:
            assert isinstance(spec, str)
            args, rep = _decode_block_str(spec)
            if args.get('num_experts', 0) > 0 and experts_multiplier > 1:
                args['num_experts'] *= experts_multiplier
            if group_size is not None:
                args.setdefault('group_size', group_size)
            stage_args.append(args)
            repeats.append(rep)
        edge_stage = stage_idx in (0, len(arch_def) - 1)
        stage_mult = 1.0 if (fix_first_last and edge_stage) else mult
        stages.append(_scale_stage_depth(stage_args, repeats, stage_mult, depth_trunc))
    return stages


# ---------------------------------------------------------------------------
# trunk builder
# ---------------------------------------------------------------------------

def get_attn_or_se(se_layer):
    if se_layer is None:
        return SqueezeExcite
    if isinstance(se_layer, str):
        if se_layer == 'se':
            return SqueezeExcite
        from ..layers import get_attn
        return get_attn(se_layer)
    return se_layer


class EfficientNetBuilder:
    """Instantiate trunk stages from decoded block args.

    Tracks running in_chs / stride / dilation across blocks, manages
    space2depth regions, and records feature_info entries at stage
    boundaries for feature-extraction wrappers.
    """

    def __init__(
            self,
            output_stride=32,
            pad_type='',
            round_chs_fn=round_channels,
            se_from_exp=False,
            act_layer=None,
            norm_layer=None,
            aa_layer=None,
            se_layer=None,
            drop_path_rate=0.,
            layer_scale_init_value=None,
            feature_location='',
    ):
        self.output_stride = output_stride
        self.pad_type = pad_type
        self.round_chs_fn = round_chs_fn
        self.se_from_exp = se_from_exp  # SE channels from expanded (mid) chs vs input
        self.act_layer = act_layer
        self.norm_layer = norm_layer
        self.aa_layer = aa_layer
        self.se_layer = get_attn_or_se(se_layer)
        try:
            self.se_layer(8, rd_ratio=1.0)  # probe rd_ratio support
            self.se_has_ratio = True
        except TypeError:
            self.se_has_ratio = False
        self.drop_path_rate = drop_path_rate
        self.layer_scale_init_value = layer_scale_init_value
        if feature_location == 'depthwise':
            _logger.warning("feature_location=='depthwise' is deprecated, using 'expansion'")
            feature_location = 'expansion'
        assert feature_location in ('bottleneck', 'expansion', '')
        self.feature_location = feature_location
        self.verbose = _DEBUG_BUILDER

        # running build state
        self.in_chs = None
        self.features = []

    # per-type constructors; ls = layer_scale_init_value
    def _factory(self, bt):
        ls = self.layer_scale_init_value
        table = {
            'ir': lambda ba: CondConvResidual(**ba) if ba.get('num_experts', 0)
                else InvertedResidual(**ba),
            'ds': DepthwiseSeparableConv,
            'dsa': DepthwiseSeparableConv,
            'er': EdgeResidual,
            'cn': ConvBnAct,
            'uir': lambda ba: UniversalInvertedResidual(**ba, layer_scale_init_value=ls),
            'mqa': lambda ba: MobileAttention(**ba, use_multi_query=True, layer_scale_init_value=ls),
            'mha': lambda ba: MobileAttention(**ba, layer_scale_init_value=ls),
        }
        ctor = table[bt]
        if isinstance(ctor, type):
            return lambda ba: ctor(**ba)
        return ctor

    def _resolve_se(self, ba, se_ratio, s2d):
        if not (se_ratio and self.se_layer is not None):
            return
        if not self.se_from_exp:
            # SE sized from block input: fold out the expansion
            se_ratio /= ba.get('exp_ratio', 1.0)
        if s2d == 1:
            se_ratio /= 4  # start of a space2depth region
        if self.se_has_ratio:
            ba['se_layer'] = partial(self.se_layer, rd_ratio=se_ratio)
        else:
            ba['se_layer'] = self.se_layer

    def _make_block(self, ba, block_idx, block_count):
        bt = ba.pop('block_type')
        ba['in_chs'] = self.in_chs
        ba['out_chs'] = self.round_chs_fn(ba['out_chs'])
        s2d = ba.get('s2d', 0)
        if s2d > 0:
            ba['out_chs'] *= 4  # channels quadruple while space2depth is active
        if ba.get('force_in_chs'):
            ba['force_in_chs'] = self.round_chs_fn(ba['force_in_chs'])
        ba['pad_type'] = self.pad_type
        ba['act_layer'] = ba['act_layer'] or self.act_layer
        assert ba['act_layer'] is not None
        ba['norm_layer'] = self.norm_layer
        ba['drop_path_rate'] = self.drop_path_rate * block_idx / block_count
        self._resolve_se(ba, ba.pop('se_ratio', None), s2d)

        _log_info_if(f'  Block type {bt} {block_idx}, Args: {ba}', self.verbose)
        block = self._factory(bt)(ba)
        self.in_chs = ba['out_chs']
        return block

    def __call__(self, in_chs, model_block_args):
        """Build all trunk stages; returns a list of nn.Sequential stages."""
        _log_info_if(f'Building model trunk with {len(model_block_args)} stages...', self.verbose)
        self.in_chs = in_chs
        block_count = sum(len(stage) for stage in model_block_args)
        block_idx = 0
        stride = 2
        dilation = 1
        s2d_phase = 0  # 0 inactive, 1 starting, 2 active
        stages = []

        if model_block_args[0][0]['stride'] > 1:
            # stem output is itself a feature level when stage 0 downsamples
            self.features.append(
                dict(module='bn1', num_chs=in_chs, stage=0, reduction=stride))

        for stage_idx, stage_args in enumerate(model_block_args):
            assert isinstance(stage_args, list)
            _log_info_if(f'Stack: {stage_idx}', self.verbose)
            blocks = []
            for i, ba in enumerate(stage_args):
                last_in_stage = i + 1 == len(stage_args)
                assert ba['stride'] in (1, 2)
                if i >= 1:
                    ba['stride'] = 1  # only a stage's first block may downsample

                # space2depth region tracking
                if not s2d_phase and ba.pop('s2d', False):
                    assert ba['stride'] == 1
                    s2d_phase = 1
                if s2d_phase > 0:
                    if s2d_phase == 2 and ba['stride'] == 2:
                        # region ends at the next downsample; undo the 4x
                        ba['stride'] = 1
                        ba['exp_ratio'] /= 4
                        s2d_phase = 0
                    else:
                        ba['s2d'] = s2d_phase

                # a stage boundary (or network end) right after this block
                # means its output is a feature level
                extract = last_in_stage and (
                    stage_idx + 1 >= len(model_block_args) or
                    model_block_args[stage_idx + 1][0]['stride'] > 1)

                # convert stride to dilation past the output_stride budget
                next_dilation = dilation
                if ba['stride'] > 1:
                    if stride * ba['stride'] > self.output_stride:
                        next_dilation = dilation * ba['stride']
                        ba['stride'] = 1
                        _log_info_if(
                            f'  Converting stride to dilation to maintain '
                            f'output_stride=={self.output_stride}', self.verbose)
                    else:
                        stride *= ba['stride']
                ba['dilation'] = dilation
                dilation = next_dilation

                blocks.append(self._make_block(ba, block_idx, block_count))
                if s2d_phase == 1:
                    s2d_phase = 2

                if extract:
                    info = dict(
                        stage=stage_idx + 1,
                        reduction=stride,
                        **blocks[-1].feature_info(self.feature_location),
                    )
                    leaf = info.get('module', '')
                    if leaf:
                        info['module'] = f'blocks.{stage_idx}.{i}.{leaf}'
                    else:
                        assert last_in_stage
                        info['module'] = f'blocks.{stage_idx}'
                    self.features.append(info)
                block_idx += 1
            stages.append(nn.Sequential(*blocks))
        return stages


# ---------------------------------------------------------------------------
# TF-style weight init
# ---------------------------------------------------------------------------

def _init_weight_goog(m, n='', fix_group_fanout=True):
    """Conv/BN/Linear init matching the TF reference implementations."""
    from ..layers.cond_conv2d import CondConv2d, get_condconv_initializer
    if isinstance(m, CondConv2d):
        fan_out = m.kernel_size[0] * m.kernel_size[1] * m.out_channels
        if fix_group_fanout:
            fan_out //= m.groups
        get_condconv_initializer(
            lambda w: nn.init.normal_(w, 0, math.sqrt(2.0 / fan_out)),
            m.num_experts, m.weight_shape)(m.weight)
        if m.bias is not None:
            nn.init.zeros_(m.bias)
    elif isinstance(m, nn.Conv2d):
        fan_out = m.kernel_size[0] * m.kernel_size[1] * m.out_channels
        if fix_group_fanout:
            fan_out //= m.groups
        nn.init.normal_(m.weight, 0, math.sqrt(2.0 / fan_out))
        if m.bias is not None:
            nn.init.zeros_(m.bias)
    elif isinstance(m, nn.BatchNorm2d):
        nn.init.ones_(m.weight)
        nn.init.zeros_(m.bias)
    elif isinstance(m, nn.Linear):
        fan_out = m.weight.size(0)
        fan_in = m.weight.size(1) if 'routing_fn' in n else 0
        bound = 1.0 / math.sqrt(fan_in + fan_out)
        nn.init.uniform_(m.weight, -bound, bound)
        if m.bias is not None:
            nn.init.zeros_(m.bias)


def efficientnet_init_weights(model: nn.Module, init_fn=None):
    init_fn = init_fn or _init_weight_goog
    for n, m in model.named_modules():
        init_fn(m, n)
