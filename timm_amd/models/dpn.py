"""DPN — Dual-Path Networks (arxiv 1707.01629).

Behavioral parity: /root/reference/timm/models/dpn.py (CatBnAct/BnActConv2d
pre-activation units, the (residual, dense) tensor pair threaded through
every block, `features.convN_i` checkpoint key layout, dpn48b..dpn131).
Stage construction here is a single loop over a width table instead of the
reference's four unrolled sections.
"""
from collections import OrderedDict
from functools import partial
from typing import Tuple, Type

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..data.constants import IMAGENET_DPN_MEAN, IMAGENET_DPN_STD
from ..layers import BatchNormAct2d, ConvNormAct, create_classifier, create_conv2d, get_norm_act_layer
from ._builder import build_model_with_cfg
from ._registry import generate_default_cfgs, register_model

__all__ = ['DPN']


class CatBnAct(nn.Module):
    """Concat the dual paths (when given a pair) then BN+act."""

    def __init__(self, in_chs: int, norm_layer: Type[nn.Module] = BatchNormAct2d):
        super().__init__()
        self.bn = norm_layer(in_chs, eps=0.001)

    def forward(self, x) -> torch.Tensor:
        return self.bn(torch.cat(x, dim=1) if isinstance(x, tuple) else x)


class BnActConv2d(nn.Module):
    """Pre-activation conv unit (BN+act before conv, DPN/DenseNet style)."""

    def __init__(
            self,
            in_chs: int,
            out_chs: int,
            kernel_size: int,
            stride: int,
            groups: int = 1,
            norm_layer: Type[nn.Module] = BatchNormAct2d,
    ):
        super().__init__()
        self.bn = norm_layer(in_chs, eps=0.001)
        self.conv = create_conv2d(in_chs, out_chs, kernel_size, stride=stride, groups=groups)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.conv(self.bn(x))


class DualPathBlock(nn.Module):
    """One DPN block: bottleneck over the concatenated paths, output split
    back into a residual-summed part and a densely-grown part."""

    def __init__(
            self,
            in_chs: int,
            num_1x1_a: int,
            num_3x3_b: int,
            num_1x1_c: int,
            inc: int,
            groups: int,
            block_type: str = 'normal',
            b: bool = False,
    ):
        super().__init__()
        assert block_type in ('proj', 'down', 'normal')
        self.num_1x1_c = num_1x1_c
        self.inc = inc
        self.b = b
        self.key_stride = 2 if block_type == 'down' else 1
        self.has_proj = block_type != 'normal'

        # stride-specific projection attribute names match converted weights
        self.c1x1_w_s1 = self.c1x1_w_s2 = None
        if self.has_proj:
            proj = BnActConv2d(
                in_chs=in_chs, out_chs=num_1x1_c + 2 * inc,
                kernel_size=1, stride=self.key_stride)
            if self.key_stride == 2:
                self.c1x1_w_s2 = proj
            else:
                self.c1x1_w_s1 = proj

        self.c1x1_a = BnActConv2d(in_chs=in_chs, out_chs=num_1x1_a, kernel_size=1, stride=1)
        self.c3x3_b = BnActConv2d(
            in_chs=num_1x1_a, out_chs=num_3x3_b, kernel_size=3,
            stride=self.key_stride, groups=groups)
        if b:
            # 'b' variants split with two separate 1x1 convs after a BN+act
            self.c1x1_c = CatBnAct(in_chs=num_3x3_b)
            self.c1x1_c1 = create_conv2d(num_3x3_b, num_1x1_c, kernel_size=1)
            self.c1x1_c2 = create_conv2d(num_3x3_b, inc, kernel_size=1)
        else:
            self.c1x1_c = BnActConv2d(
                in_chs=num_3x3_b, out_chs=num_1x1_c + inc, kernel_size=1, stride=1)
            self.c1x1_c1 = self.c1x1_c2 = None

    def _shortcut(self, x, x_cat):
        """Resolve the (residual, dense) shortcut pair for this block."""
        if not self.has_proj:
            return x[0], x[1]
        proj = self.c1x1_w_s2 if self.c1x1_w_s2 is not None else self.c1x1_w_s1
        projected = proj(x_cat)
        return projected[:, :self.num_1x1_c], projected[:, self.num_1x1_c:]

    def forward(self, x) -> Tuple[torch.Tensor, torch.Tensor]:
        x_cat = torch.cat(x, dim=1) if isinstance(x, tuple) else x
        short_res, short_dense = self._shortcut(x, x_cat)

        out = self.c1x1_c(self.c3x3_b(self.c1x1_a(x_cat)))
        if self.c1x1_c1 is not None:
            res_part, dense_part = self.c1x1_c1(out), self.c1x1_c2(out)
        else:
            res_part, dense_part = out[:, :self.num_1x1_c], out[:, self.num_1x1_c:]
        return short_res + res_part, torch.cat([short_dense, dense_part], dim=1)


class DPN(nn.Module):
    """Dual-Path Network trunk + 1x1-conv classifier head."""

    # per-stage (base width multiplier, reduction) — widths scale by bw_factor
    _STAGES = ((64, 4), (128, 8), (256, 16), (512, 32))

    def __init__(
            self,
            k_sec: Tuple[int, ...] = (3, 4, 20, 3),
            inc_sec: Tuple[int, ...] = (16, 32, 24, 128),
            k_r: int = 96,
            groups: int = 32,
            num_classes: int = 1000,
            in_chans: int = 3,
            output_stride: int = 32,
            global_pool: str = 'avg',
            small: bool = False,
            num_init_features: int = 64,
            b: bool = False,
            drop_rate: float = 0.,
            norm_layer: str = 'batchnorm2d',
            act_layer: str = 'relu',
            fc_act_layer: str = 'elu',
    ):
        super().__init__()
        assert output_stride == 32
        self.num_classes = num_classes
        self.drop_rate = drop_rate
        self.b = b

        norm_layer = partial(get_norm_act_layer(norm_layer, act_layer=act_layer), eps=.001)
        fc_norm_layer = partial(
            get_norm_act_layer(norm_layer, act_layer=fc_act_layer), eps=.001, inplace=False)
        bw_factor = 1 if small else 4

        blocks = OrderedDict()
        blocks['conv1_1'] = ConvNormAct(
            in_chans, num_init_features,
            kernel_size=3 if small else 7, stride=2, norm_layer=norm_layer)
        blocks['conv1_pool'] = nn.MaxPool2d(kernel_size=3, stride=2, padding=1)
        self.feature_info = [
            dict(num_chs=num_init_features, reduction=2, module='features.conv1_1')]

        in_chs = num_init_features
        for stage_idx, ((base_bw, reduction), depth, inc) in enumerate(
                zip(self._STAGES, k_sec, inc_sec)):
            bw = base_bw * bw_factor
            r = (k_r * bw) // (64 * bw_factor)
            name = f'conv{stage_idx + 2}'
            first_type = 'proj' if stage_idx == 0 else 'down'
            blocks[f'{name}_1'] = DualPathBlock(in_chs, r, r, bw, inc, groups, first_type, b)
            in_chs = bw + 3 * inc
            for i in range(2, depth + 1):
                blocks[f'{name}_{i}'] = DualPathBlock(in_chs, r, r, bw, inc, groups, 'normal', b)
                in_chs += inc
            self.feature_info += [
                dict(num_chs=in_chs, reduction=reduction * 2, module=f'features.{name}_{depth}')]

        blocks['conv5_bn_ac'] = CatBnAct(in_chs, norm_layer=fc_norm_layer)
        self.num_features = self.head_hidden_size = in_chs
        self.features = nn.Sequential(blocks)

        # conv-style classifier allows test-time spatial pooling of logits
        self.global_pool, self.classifier = create_classifier(
            self.num_features, self.num_classes, pool_type=global_pool, use_conv=True)
        self.flatten = nn.Flatten(1) if global_pool else nn.Identity()

    @torch.jit.ignore
    def group_matcher(self, coarse: bool = False):
        return dict(
            stem=r'^features\.conv1',
            blocks=[
                (r'^features\.conv(\d+)' if coarse else r'^features\.conv(\d+)_(\d+)', None),
                (r'^features\.conv5_bn_ac', (99999,)),
            ],
        )

    @torch.jit.ignore
    def set_grad_checkpointing(self, enable: bool = True):
        assert not enable, 'gradient checkpointing not supported'

    @torch.jit.ignore
    def get_classifier(self) -> nn.Module:
        return self.classifier

    def reset_classifier(self, num_classes: int, global_pool: str = 'avg'):
        self.num_classes = num_classes
        self.global_pool, self.classifier = create_classifier(
            self.num_features, self.num_classes, pool_type=global_pool, use_conv=True)
        self.flatten = nn.Flatten(1) if global_pool else nn.Identity()

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        return self.features(x)

    def forward_head(self, x: torch.Tensor, pre_logits: bool = False) -> torch.Tensor:
        x = self.global_pool(x)
        if self.drop_rate > 0.:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        if pre_logits:
            return self.flatten(x)
        return self.flatten(self.classifier(x))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.forward_head(self.forward_features(x))


def _create_dpn(variant, pretrained=False, **kwargs):
    return build_model_with_cfg(
        DPN, variant, pretrained,
        feature_cfg=dict(feature_concat=True, flatten_sequential=True),
        **kwargs,
    )


def _cfg(url='', **kwargs):
    return {
        'url': url, 'num_classes': 1000, 'input_size': (3, 224, 224), 'pool_size': (7, 7),
        'crop_pct': 0.875, 'interpolation': 'bicubic',
        'mean': IMAGENET_DPN_MEAN, 'std': IMAGENET_DPN_STD,
        'first_conv': 'features.conv1_1.conv', 'classifier': 'classifier',
        **kwargs,
    }


default_cfgs = generate_default_cfgs({
    'dpn48b.untrained': _cfg(),
    'dpn68.mx_in1k': _cfg(),
    'dpn68b.ra_in1k': _cfg(test_input_size=(3, 288, 288), test_crop_pct=1.0),
    'dpn92.mx_in1k': _cfg(),
    'dpn98.mx_in1k': _cfg(),
    'dpn131.mx_in1k': _cfg(),
    'dpn107.mx_in1k': _cfg(),
})

# variant name -> constructor args (widths/depths from the paper releases)
_VARIANTS = dict(
    dpn48b=dict(
        small=True, num_init_features=10, k_r=128, groups=32,
        b=True, k_sec=(3, 4, 6, 3), inc_sec=(16, 32, 32, 64), act_layer='silu'),
    dpn68=dict(
        small=True, num_init_features=10, k_r=128, groups=32,
        k_sec=(3, 4, 12, 3), inc_sec=(16, 32, 32, 64)),
    dpn68b=dict(
        small=True, num_init_features=10, k_r=128, groups=32,
        b=True, k_sec=(3, 4, 12, 3), inc_sec=(16, 32, 32, 64)),
    dpn92=dict(
        num_init_features=64, k_r=96, groups=32,
        k_sec=(3, 4, 20, 3), inc_sec=(16, 32, 24, 128)),
    dpn98=dict(
        num_init_features=96, k_r=160, groups=40,
        k_sec=(3, 6, 20, 3), inc_sec=(16, 32, 32, 128)),
    dpn131=dict(
        num_init_features=128, k_r=160, groups=40,
        k_sec=(4, 8, 28, 3), inc_sec=(16, 32, 32, 128)),
    dpn107=dict(
        num_init_features=128, k_r=200, groups=50,
        k_sec=(4, 8, 20, 3), inc_sec=(20, 64, 64, 128)),
)


@register_model
def dpn48b(pretrained=False, **kwargs) -> DPN:
    return _create_dpn('dpn48b', pretrained=pretrained, **dict(_VARIANTS['dpn48b'], **kwargs))


@register_model
def dpn68(pretrained=False, **kwargs) -> DPN:
    return _create_dpn('dpn68', pretrained=pretrained, **dict(_VARIANTS['dpn68'], **kwargs))


@register_model
def dpn68b(pretrained=False, **kwargs) -> DPN:
    return _create_dpn('dpn68b', pretrained=pretrained, **dict(_VARIANTS['dpn68b'], **kwargs))


@register_model
def dpn92(pretrained=False, **kwargs) -> DPN:
    return _create_dpn('dpn92', pretrained=pretrained, **dict(_VARIANTS['dpn92'], **kwargs))


@register_model
def dpn98(pretrained=False, **kwargs) -> DPN:
    return _create_dpn('dpn98', pretrained=pretrained, **dict(_VARIANTS['dpn98'], **kwargs))


@register_model
def dpn131(pretrained=False, **kwargs) -> DPN:
    return _create_dpn('dpn131', pretrained=pretrained, **dict(_VARIANTS['dpn131'], **kwargs))


@register_model
def dpn107(pretrained=False, **kwargs) -> DPN:
    return _create_dpn('dpn107', pretrained=pretrained, **dict(_VARIANTS['dpn107'], **kwargs))
