"""Bring-Your-Own-Attention networks (BoTNet, HaloNet, LambdaNet hybrids).

Capability parity with reference `timm/models/byoanet.py`: ByobNet configs
mixing conv bottleneck blocks with self-attention blocks — bottleneck
(global 2D rel-pos) attention, halo (blocked local) attention and lambda
layers — via the `self_attn` block type (see byobnet.py / layers/conv_self_attn.py).

All attention inner loops are batched GEMMs + softmax; on MI355X they run on
hipBLASLt strided-batched GEMM with our fused softmax where applicable.
"""
from typing import Any, Dict, Optional

from ..data.constants import IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD
from ._builder import build_model_with_cfg
from ._registry import generate_default_cfgs, register_model
from .byobnet import ByoBlockCfg, ByoModelCfg, ByobNet, interleave_blocks

__all__ = []


def _tiered_26t_blocks(self_attn_c3=True, gs=0, br=0.25):
    """Shared ResNet26-T layout: interleaved self-attn in stage 3, full self-attn stage 4."""
    return (
        ByoBlockCfg(type='bottle', d=2, c=256, s=1, gs=gs, br=br),
        ByoBlockCfg(type='bottle', d=2, c=512, s=2, gs=gs, br=br),
        interleave_blocks(types=('bottle', 'self_attn'), d=2, c=1024, s=2, gs=gs, br=br),
        ByoBlockCfg(type='self_attn', d=2, c=2048, s=2, gs=gs, br=br),
    )


def _50ts_blocks(**stage2_kw):
    """Shared ResNet50-T layout w/ silu: self-attn interleaved in stages 2-4."""
    return (
        ByoBlockCfg(type='bottle', d=3, c=256, s=1, gs=0, br=0.25),
        interleave_blocks(types=('bottle', 'self_attn'), every=4, d=4, c=512, s=2, gs=0, br=0.25, **stage2_kw),
        interleave_blocks(types=('bottle', 'self_attn'), d=6, c=1024, s=2, gs=0, br=0.25),
        interleave_blocks(types=('bottle', 'self_attn'), d=3, c=2048, s=2, gs=0, br=0.25),
    )


def _33ts_blocks():
    """ResNet33-T layout, one self-attn at end of stages 2/3, self-attn stage 4."""
    return (
        ByoBlockCfg(type='bottle', d=2, c=256, s=1, gs=0, br=0.25),
        interleave_blocks(types=('bottle', 'self_attn'), every=[2], d=3, c=512, s=2, gs=0, br=0.25),
        interleave_blocks(types=('bottle', 'self_attn'), every=[2], d=3, c=1024, s=2, gs=0, br=0.25),
        ByoBlockCfg('self_attn', d=2, c=1536, s=2, gs=0, br=0.333),
    )


model_cfgs = dict(
    botnet26t=ByoModelCfg(
        blocks=_tiered_26t_blocks(),
        stem_chs=64, stem_type='tiered', stem_pool='maxpool',
        fixed_input_size=True,
        self_attn_layer='bottleneck', self_attn_kwargs=dict(),
    ),
    sebotnet33ts=ByoModelCfg(
        blocks=_33ts_blocks(),
        stem_chs=64, stem_type='tiered', stem_pool='',
        act_layer='silu', num_features=1280, attn_layer='se',
        self_attn_layer='bottleneck', self_attn_kwargs=dict(),
    ),
    botnet50ts=ByoModelCfg(
        blocks=_50ts_blocks(),
        stem_chs=64, stem_type='tiered', stem_pool='maxpool',
        act_layer='silu', fixed_input_size=True,
        self_attn_layer='bottleneck', self_attn_kwargs=dict(),
    ),
    eca_botnext26ts=ByoModelCfg(
        blocks=_tiered_26t_blocks(gs=16),
        stem_chs=64, stem_type='tiered', stem_pool='maxpool',
        fixed_input_size=True, act_layer='silu', attn_layer='eca',
        self_attn_layer='bottleneck', self_attn_kwargs=dict(dim_head=16),
    ),

    halonet_h1=ByoModelCfg(
        blocks=(
            ByoBlockCfg(type='self_attn', d=3, c=64, s=1, gs=0, br=1.0),
            ByoBlockCfg(type='self_attn', d=3, c=128, s=2, gs=0, br=1.0),
            ByoBlockCfg(type='self_attn', d=10, c=256, s=2, gs=0, br=1.0),
            ByoBlockCfg(type='self_attn', d=3, c=512, s=2, gs=0, br=1.0),
        ),
        stem_chs=64, stem_type='7x7', stem_pool='maxpool',
        self_attn_layer='halo', self_attn_kwargs=dict(block_size=8, halo_size=3),
    ),
    halonet26t=ByoModelCfg(
        blocks=_tiered_26t_blocks(),
        stem_chs=64, stem_type='tiered', stem_pool='maxpool',
        self_attn_layer='halo', self_attn_kwargs=dict(block_size=8, halo_size=2),
    ),
    sehalonet33ts=ByoModelCfg(
        blocks=_33ts_blocks(),
        stem_chs=64, stem_type='tiered', stem_pool='',
        act_layer='silu', num_features=1280, attn_layer='se',
        self_attn_layer='halo', self_attn_kwargs=dict(block_size=8, halo_size=3),
    ),
    halonet50ts=ByoModelCfg(
        blocks=_50ts_blocks(
            self_attn_layer='halo',
            self_attn_kwargs=dict(block_size=8, halo_size=3, num_heads=4),
        ),
        stem_chs=64, stem_type='tiered', stem_pool='maxpool',
        act_layer='silu',
        self_attn_layer='halo', self_attn_kwargs=dict(block_size=8, halo_size=3),
    ),
    eca_halonext26ts=ByoModelCfg(
        blocks=_tiered_26t_blocks(gs=16),
        stem_chs=64, stem_type='tiered', stem_pool='maxpool',
        act_layer='silu', attn_layer='eca',
        self_attn_layer='halo', self_attn_kwargs=dict(block_size=8, halo_size=2, dim_head=16),
    ),

    lambda_resnet26t=ByoModelCfg(
        blocks=_tiered_26t_blocks(),
        stem_chs=64, stem_type='tiered', stem_pool='maxpool',
        self_attn_layer='lambda', self_attn_kwargs=dict(r=9),
    ),
    lambda_resnet50ts=ByoModelCfg(
        blocks=_50ts_blocks(),
        stem_chs=64, stem_type='tiered', stem_pool='maxpool',
        act_layer='silu',
        self_attn_layer='lambda', self_attn_kwargs=dict(r=9),
    ),
    lambda_resnet26rpt_256=ByoModelCfg(
        blocks=_tiered_26t_blocks(),
        stem_chs=64, stem_type='tiered', stem_pool='maxpool',
        self_attn_layer='lambda', self_attn_kwargs=dict(r=None),
    ),

    haloregnetz_b=ByoModelCfg(
        blocks=(
            ByoBlockCfg(type='bottle', d=2, c=48, s=2, gs=16, br=3),
            ByoBlockCfg(type='bottle', d=6, c=96, s=2, gs=16, br=3),
            interleave_blocks(types=('bottle', 'self_attn'), every=3, d=12, c=192, s=2, gs=16, br=3),
            ByoBlockCfg('self_attn', d=2, c=288, s=2, gs=16, br=3),
        ),
        stem_chs=32, stem_pool='', downsample='',
        num_features=1536, act_layer='silu',
        attn_layer='se', attn_kwargs=dict(rd_ratio=0.25),
        block_kwargs=dict(bottle_in=True, linear_out=True),
        self_attn_layer='halo', self_attn_kwargs=dict(block_size=7, halo_size=2, qk_ratio=0.33),
    ),

    lamhalobotnet50ts=ByoModelCfg(
        blocks=(
            ByoBlockCfg(type='bottle', d=3, c=256, s=1, gs=0, br=0.25),
            interleave_blocks(
                types=('bottle', 'self_attn'), d=4, c=512, s=2, gs=0, br=0.25,
                self_attn_layer='lambda', self_attn_kwargs=dict(r=13)),
            interleave_blocks(
                types=('bottle', 'self_attn'), d=6, c=1024, s=2, gs=0, br=0.25,
                self_attn_layer='halo', self_attn_kwargs=dict(halo_size=3)),
            interleave_blocks(
                types=('bottle', 'self_attn'), d=3, c=2048, s=2, gs=0, br=0.25,
                self_attn_layer='bottleneck', self_attn_kwargs=dict()),
        ),
        stem_chs=64, stem_type='tiered', stem_pool='',
        act_layer='silu',
    ),
    halo2botnet50ts=ByoModelCfg(
        blocks=(
            ByoBlockCfg(type='bottle', d=3, c=256, s=1, gs=0, br=0.25),
            interleave_blocks(
                types=('bottle', 'self_attn'), d=4, c=512, s=2, gs=0, br=0.25,
                self_attn_layer='halo', self_attn_kwargs=dict(halo_size=3)),
            interleave_blocks(
                types=('bottle', 'self_attn'), d=6, c=1024, s=2, gs=0, br=0.25,
                self_attn_layer='halo', self_attn_kwargs=dict(halo_size=3)),
            interleave_blocks(
                types=('bottle', 'self_attn'), d=3, c=2048, s=2, gs=0, br=0.25,
                self_attn_layer='bottleneck', self_attn_kwargs=dict()),
        ),
        stem_chs=64, stem_type='tiered', stem_pool='',
        act_layer='silu',
    ),
)


def _create_byoanet(variant: str, cfg_variant: Optional[str] = None, pretrained: bool = False, **kwargs) -> ByobNet:
    return build_model_with_cfg(
        ByobNet, variant, pretrained,
        model_cfg=model_cfgs[cfg_variant or variant],
        feature_cfg=dict(flatten_sequential=True),
        **kwargs,
    )


def _cfg(url: str = '', **kwargs) -> Dict[str, Any]:
    return {
        'url': url, 'num_classes': 1000, 'input_size': (3, 224, 224), 'pool_size': (7, 7),
        'crop_pct': 0.95, 'interpolation': 'bicubic',
        'mean': IMAGENET_DEFAULT_MEAN, 'std': IMAGENET_DEFAULT_STD,
        'first_conv': 'stem.conv1.conv', 'classifier': 'head.fc',
        'fixed_input_size': False, 'min_input_size': (3, 224, 224),
        **kwargs
    }


default_cfgs = generate_default_cfgs({
    'botnet26t_256.c1_in1k': _cfg(
        hf_hub_id='timm/', fixed_input_size=True, input_size=(3, 256, 256), pool_size=(8, 8)),
    'sebotnet33ts_256.a1h_in1k': _cfg(
        hf_hub_id='timm/', fixed_input_size=True, input_size=(3, 256, 256), pool_size=(8, 8), crop_pct=0.94),
    'botnet50ts_256.untrained': _cfg(
        fixed_input_size=True, input_size=(3, 256, 256), pool_size=(8, 8)),
    'eca_botnext26ts_256.c1_in1k': _cfg(
        hf_hub_id='timm/', fixed_input_size=True, input_size=(3, 256, 256), pool_size=(8, 8)),

    'halonet_h1.untrained': _cfg(input_size=(3, 256, 256), pool_size=(8, 8), min_input_size=(3, 256, 256)),
    'halonet26t.a1h_in1k': _cfg(
        hf_hub_id='timm/', input_size=(3, 256, 256), pool_size=(8, 8), min_input_size=(3, 256, 256)),
    'sehalonet33ts.ra2_in1k': _cfg(
        hf_hub_id='timm/', input_size=(3, 256, 256), pool_size=(8, 8), min_input_size=(3, 256, 256), crop_pct=0.94),
    'halonet50ts.a1h_in1k': _cfg(
        hf_hub_id='timm/', input_size=(3, 256, 256), pool_size=(8, 8), min_input_size=(3, 256, 256), crop_pct=0.94),
    'eca_halonext26ts.c1_in1k': _cfg(
        hf_hub_id='timm/', input_size=(3, 256, 256), pool_size=(8, 8), min_input_size=(3, 256, 256), crop_pct=0.94),

    'lambda_resnet26t.c1_in1k': _cfg(
        hf_hub_id='timm/', min_input_size=(3, 128, 128), input_size=(3, 256, 256), pool_size=(8, 8), crop_pct=0.94),
    'lambda_resnet50ts.a1h_in1k': _cfg(
        hf_hub_id='timm/', min_input_size=(3, 128, 128), input_size=(3, 256, 256), pool_size=(8, 8)),
    'lambda_resnet26rpt_256.c1_in1k': _cfg(
        hf_hub_id='timm/', fixed_input_size=True, input_size=(3, 256, 256), pool_size=(8, 8), crop_pct=0.94),

    'haloregnetz_b.ra3_in1k': _cfg(
        hf_hub_id='timm/', mean=(0.5, 0.5, 0.5), std=(0.5, 0.5, 0.5),
        first_conv='stem.conv', input_size=(3, 224, 224), pool_size=(7, 7),
        min_input_size=(3, 224, 224), crop_pct=0.94),

    'lamhalobotnet50ts_256.a1h_in1k': _cfg(
        hf_hub_id='timm/', fixed_input_size=True, input_size=(3, 256, 256), pool_size=(8, 8)),
    'halo2botnet50ts_256.a1h_in1k': _cfg(
        hf_hub_id='timm/', fixed_input_size=True, input_size=(3, 256, 256), pool_size=(8, 8)),
})


@register_model
def botnet26t_256(pretrained: bool = False, **kwargs) -> ByobNet:
    """Bottleneck Transformer w/ ResNet26-T backbone."""
    kwargs.setdefault('img_size', 256)
    return _create_byoanet('botnet26t_256', 'botnet26t', pretrained=pretrained, **kwargs)


@register_model
def sebotnet33ts_256(pretrained: bool = False, **kwargs) -> ByobNet:
    """Bottleneck Transformer w/ ResNet33-T backbone, SE attn, SiLU."""
    return _create_byoanet('sebotnet33ts_256', 'sebotnet33ts', pretrained=pretrained, **kwargs)


@register_model
def botnet50ts_256(pretrained: bool = False, **kwargs) -> ByobNet:
    """Bottleneck Transformer w/ ResNet50-T backbone, SiLU."""
    kwargs.setdefault('img_size', 256)
    return _create_byoanet('botnet50ts_256', 'botnet50ts', pretrained=pretrained, **kwargs)


@register_model
def eca_botnext26ts_256(pretrained: bool = False, **kwargs) -> ByobNet:
    """Bottleneck Transformer w/ ResNeXt26-T backbone, ECA, SiLU."""
    kwargs.setdefault('img_size', 256)
    return _create_byoanet('eca_botnext26ts_256', 'eca_botnext26ts', pretrained=pretrained, **kwargs)


@register_model
def halonet_h1(pretrained: bool = False, **kwargs) -> ByobNet:
    """HaloNet-H1: halo attention in every stage (slow)."""
    return _create_byoanet('halonet_h1', pretrained=pretrained, **kwargs)


@register_model
def halonet26t(pretrained: bool = False, **kwargs) -> ByobNet:
    """HaloNet w/ ResNet26-T backbone, halo attn in final two stages."""
    return _create_byoanet('halonet26t', pretrained=pretrained, **kwargs)


@register_model
def sehalonet33ts(pretrained: bool = False, **kwargs) -> ByobNet:
    """HaloNet w/ ResNet33-T backbone, SE attn, SiLU."""
    return _create_byoanet('sehalonet33ts', pretrained=pretrained, **kwargs)


@register_model
def halonet50ts(pretrained: bool = False, **kwargs) -> ByobNet:
    """HaloNet w/ ResNet50-T backbone, SiLU."""
    return _create_byoanet('halonet50ts', pretrained=pretrained, **kwargs)


@register_model
def eca_halonext26ts(pretrained: bool = False, **kwargs) -> ByobNet:
    """HaloNet w/ ResNeXt26-T backbone, ECA, SiLU."""
    return _create_byoanet('eca_halonext26ts', pretrained=pretrained, **kwargs)


@register_model
def lambda_resnet26t(pretrained: bool = False, **kwargs) -> ByobNet:
    """Lambda-ResNet26-T, conv-pos lambda layers in last two stages."""
    return _create_byoanet('lambda_resnet26t', pretrained=pretrained, **kwargs)


@register_model
def lambda_resnet50ts(pretrained: bool = False, **kwargs) -> ByobNet:
    """Lambda-ResNet50-TS, SiLU, conv-pos lambda layers."""
    return _create_byoanet('lambda_resnet50ts', pretrained=pretrained, **kwargs)


@register_model
def lambda_resnet26rpt_256(pretrained: bool = False, **kwargs) -> ByobNet:
    """Lambda-ResNet26-R-T, rel-pos-embed lambda layers."""
    kwargs.setdefault('img_size', 256)
    return _create_byoanet('lambda_resnet26rpt_256', pretrained=pretrained, **kwargs)


@register_model
def haloregnetz_b(pretrained: bool = False, **kwargs) -> ByobNet:
    """Halo attention + RegNetZ-B backbone."""
    return _create_byoanet('haloregnetz_b', pretrained=pretrained, **kwargs)


@register_model
def lamhalobotnet50ts_256(pretrained: bool = False, **kwargs) -> ByobNet:
    """Combo: lambda (stage 2) + halo (stage 3) + bottleneck (stage 4) attention."""
    return _create_byoanet('lamhalobotnet50ts_256', 'lamhalobotnet50ts', pretrained=pretrained, **kwargs)


@register_model
def halo2botnet50ts_256(pretrained: bool = False, **kwargs) -> ByobNet:
    """Combo: halo (stages 2-3) + bottleneck (stage 4) attention."""
    return _create_byoanet('halo2botnet50ts_256', 'halo2botnet50ts', pretrained=pretrained, **kwargs)
