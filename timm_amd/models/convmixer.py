"""ConvMixer ("Patches Are All You Need?", arxiv 2201.09792).

Behavioral parity: /root/reference/timm/models/convmixer.py (patch-conv stem,
depth x [residual depthwise + pointwise] mixer blocks, checkpoint key layout
via nn.Sequential indices).
"""
from typing import Optional, Type

import torch
import torch.nn as nn

from ..data.constants import IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD
from ..layers import SelectAdaptivePool2d
from ._builder import build_model_with_cfg
from ._manipulate import checkpoint_seq
from ._registry import generate_default_cfgs, register_model

__all__ = ['ConvMixer']


class Residual(nn.Module):
    def __init__(self, fn: nn.Module):
        super().__init__()
        self.fn = fn

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.fn(x) + x


def _conv_act_bn(conv: nn.Conv2d, act_layer: Type[nn.Module]) -> list:
    """The repeating conv -> act -> BN triple (BN after act, ConvMixer style)."""
    return [conv, act_layer(), nn.BatchNorm2d(conv.out_channels)]


def _mixer_block(dim: int, kernel_size: int, act_layer: Type[nn.Module]) -> nn.Sequential:
    """One block: residual 'same'-padded depthwise mix + pointwise channel mix."""
    spatial_mix = Residual(nn.Sequential(
        *_conv_act_bn(nn.Conv2d(dim, dim, kernel_size, groups=dim, padding='same'), act_layer)))
    channel_mix = _conv_act_bn(nn.Conv2d(dim, dim, kernel_size=1), act_layer)
    return nn.Sequential(spatial_mix, *channel_mix)


class ConvMixer(nn.Module):
    """Isotropic conv mixer over patch embeddings."""

    def __init__(
            self,
            dim: int,
            depth: int,
            kernel_size: int = 9,
            patch_size: int = 7,
            in_chans: int = 3,
            num_classes: int = 1000,
            global_pool: str = 'avg',
            drop_rate: float = 0.,
            act_layer: Type[nn.Module] = nn.GELU,
            **kwargs,
    ):
        super().__init__()
        self.num_classes = num_classes
        self.num_features = self.head_hidden_size = dim
        self.grad_checkpointing = False

        self.stem = nn.Sequential(
            *_conv_act_bn(
                nn.Conv2d(in_chans, dim, kernel_size=patch_size, stride=patch_size), act_layer))
        self.blocks = nn.Sequential(
            *[_mixer_block(dim, kernel_size, act_layer) for _ in range(depth)])
        self.pooling = SelectAdaptivePool2d(pool_type=global_pool, flatten=True)
        self.head_drop = nn.Dropout(drop_rate)
        self.head = nn.Linear(dim, num_classes) if num_classes > 0 else nn.Identity()

    @torch.jit.ignore
    def group_matcher(self, coarse: bool = False):
        return dict(stem=r'^stem', blocks=r'^blocks\.(\d+)')

    @torch.jit.ignore
    def set_grad_checkpointing(self, enable: bool = True):
        self.grad_checkpointing = enable

    @torch.jit.ignore
    def get_classifier(self) -> nn.Module:
        return self.head

    def reset_classifier(self, num_classes: int, global_pool: Optional[str] = None):
        self.num_classes = num_classes
        if global_pool is not None:
            self.pooling = SelectAdaptivePool2d(pool_type=global_pool, flatten=True)
        self.head = nn.Linear(self.num_features, num_classes) if num_classes > 0 else nn.Identity()

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        x = self.stem(x)
        if self.grad_checkpointing and not torch.jit.is_scripting():
            return checkpoint_seq(self.blocks, x)
        return self.blocks(x)

    def forward_head(self, x: torch.Tensor, pre_logits: bool = False) -> torch.Tensor:
        x = self.head_drop(self.pooling(x))
        return x if pre_logits else self.head(x)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.forward_head(self.forward_features(x))


def _create_convmixer(variant, pretrained=False, **kwargs):
    return build_model_with_cfg(ConvMixer, variant, pretrained, **kwargs)


def _cfg(url='', **kwargs):
    return {
        'url': url, 'num_classes': 1000, 'input_size': (3, 224, 224), 'pool_size': None,
        'crop_pct': .96, 'interpolation': 'bicubic',
        'mean': IMAGENET_DEFAULT_MEAN, 'std': IMAGENET_DEFAULT_STD,
        'first_conv': 'stem.0', 'classifier': 'head',
        **kwargs,
    }


default_cfgs = generate_default_cfgs({
    'convmixer_1536_20.in1k': _cfg(),
    'convmixer_768_32.in1k': _cfg(),
    'convmixer_1024_20_ks9_p14.in1k': _cfg(),
})


@register_model
def convmixer_1536_20(pretrained=False, **kwargs) -> ConvMixer:
    args = dict(dim=1536, depth=20, kernel_size=9, patch_size=7)
    return _create_convmixer('convmixer_1536_20', pretrained, **dict(args, **kwargs))


@register_model
def convmixer_768_32(pretrained=False, **kwargs) -> ConvMixer:
    args = dict(dim=768, depth=32, kernel_size=7, patch_size=7, act_layer=nn.ReLU)
    return _create_convmixer('convmixer_768_32', pretrained, **dict(args, **kwargs))


@register_model
def convmixer_1024_20_ks9_p14(pretrained=False, **kwargs) -> ConvMixer:
    args = dict(dim=1024, depth=20, kernel_size=9, patch_size=14)
    return _create_convmixer('convmixer_1024_20_ks9_p14', pretrained, **dict(args, **kwargs))
