"""Legacy SENet / SE-ResNet / SE-ResNeXt — MI355X-native implementation.

Capability parity with reference `timm/models/senet.py` (Cadene-lineage
topology): `SEModule` (:36), Caffe-style SE bottlenecks with stride on conv1
(`SEResNetBottleneck` :122), SENet154's double-width bottleneck (:83),
SE-ResNeXt bottleneck (:153) and basic block (:185).
"""
import math
from collections import OrderedDict
from typing import Optional, Tuple, Type

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..data.constants import IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD
from ..layers import create_classifier
from ._builder import build_model_with_cfg
from ._registry import generate_default_cfgs, register_model

__all__ = ['SENet']


def _weight_init(m):
    if isinstance(m, nn.Conv2d):
        nn.init.kaiming_normal_(m.weight, mode='fan_out', nonlinearity='relu')
    elif isinstance(m, nn.BatchNorm2d):
        nn.init.ones_(m.weight)
        nn.init.zeros_(m.bias)


class SEModule(nn.Module):
    def __init__(self, channels: int, reduction: int):
        super().__init__()
        self.fc1 = nn.Conv2d(channels, channels // reduction, kernel_size=1)
        self.relu = nn.ReLU(inplace=True)
        self.fc2 = nn.Conv2d(channels // reduction, channels, kernel_size=1)
        self.sigmoid = nn.Sigmoid()

    def forward(self, x):
        module_input = x
        x = x.mean((2, 3), keepdim=True)
        x = self.relu(self.fc1(x))
        x = self.sigmoid(self.fc2(x))
        return module_input * x


class Bottleneck(nn.Module):
    """Base bottleneck with conv1/bn1..conv3/bn3 attrs + SE before the add."""

    def forward(self, x):
        shortcut = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            shortcut = self.downsample(x)
        out = self.se_module(out) + shortcut
        return self.relu(out)


class SEBottleneck(Bottleneck):
    """SENet154 bottleneck: double-width 1x1 then grouped 3x3."""
    expansion = 4

    def __init__(self, inplanes, planes, groups, reduction, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes * 2, kernel_size=1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes * 2)
        self.conv2 = nn.Conv2d(
            planes * 2, planes * 4, kernel_size=3, stride=stride, padding=1, groups=groups, bias=False)
        self.bn2 = nn.BatchNorm2d(planes * 4)
        self.conv3 = nn.Conv2d(planes * 4, planes * 4, kernel_size=1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * 4)
        self.relu = nn.ReLU(inplace=True)
        self.se_module = SEModule(planes * 4, reduction=reduction)
        self.downsample = downsample
        self.stride = stride


class SEResNetBottleneck(Bottleneck):
    """Caffe-style SE-ResNet bottleneck: stride on conv1 (not conv2)."""
    expansion = 4

    def __init__(self, inplanes, planes, groups, reduction, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, kernel_size=1, bias=False, stride=stride)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, kernel_size=3, padding=1, groups=groups, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv3 = nn.Conv2d(planes, planes * 4, kernel_size=1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * 4)
        self.relu = nn.ReLU(inplace=True)
        self.se_module = SEModule(planes * 4, reduction=reduction)
        self.downsample = downsample
        self.stride = stride


class SEResNeXtBottleneck(Bottleneck):
    """SE-ResNeXt bottleneck type C."""
    expansion = 4

    def __init__(self, inplanes, planes, groups, reduction, stride=1, downsample=None, base_width=4):
        super().__init__()
        width = math.floor(planes * (base_width / 64)) * groups
        self.conv1 = nn.Conv2d(inplanes, width, kernel_size=1, bias=False, stride=1)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, kernel_size=3, stride=stride, padding=1, groups=groups, bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, planes * 4, kernel_size=1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * 4)
        self.relu = nn.ReLU(inplace=True)
        self.se_module = SEModule(planes * 4, reduction=reduction)
        self.downsample = downsample
        self.stride = stride


class SEResNetBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, groups, reduction, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, kernel_size=3, padding=1, stride=stride, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, kernel_size=3, padding=1, groups=groups, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.relu = nn.ReLU(inplace=True)
        self.se_module = SEModule(planes, reduction=reduction)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        shortcut = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        if self.downsample is not None:
            shortcut = self.downsample(x)
        out = self.se_module(out) + shortcut
        return self.relu(out)


class SENet(nn.Module):
    """Legacy SENet (reference `senet.py:229`)."""

    def __init__(
            self,
            block: Type[nn.Module],
            layers: Tuple[int, ...],
            groups: int,
            reduction: int,
            drop_rate: float = 0.2,
            in_chans: int = 3,
            inplanes: int = 64,
            input_3x3: bool = False,
            downsample_kernel_size: int = 1,
            downsample_padding: int = 0,
            num_classes: int = 1000,
            global_pool: str = 'avg',
    ):
        super().__init__()
        self.inplanes = inplanes
        self.num_classes = num_classes
        self.drop_rate = drop_rate
        if input_3x3:
            layer0_modules = [
                ('conv1', nn.Conv2d(in_chans, 64, 3, stride=2, padding=1, bias=False)),
                ('bn1', nn.BatchNorm2d(64)),
                ('relu1', nn.ReLU(inplace=True)),
                ('conv2', nn.Conv2d(64, 64, 3, stride=1, padding=1, bias=False)),
                ('bn2', nn.BatchNorm2d(64)),
                ('relu2', nn.ReLU(inplace=True)),
                ('conv3', nn.Conv2d(64, inplanes, 3, stride=1, padding=1, bias=False)),
                ('bn3', nn.BatchNorm2d(inplanes)),
                ('relu3', nn.ReLU(inplace=True)),
            ]
        else:
            layer0_modules = [
                ('conv1', nn.Conv2d(in_chans, inplanes, kernel_size=7, stride=2, padding=3, bias=False)),
                ('bn1', nn.BatchNorm2d(inplanes)),
                ('relu1', nn.ReLU(inplace=True)),
            ]
        self.layer0 = nn.Sequential(OrderedDict(layer0_modules))
        self.pool0 = nn.MaxPool2d(3, stride=2, ceil_mode=True)
        self.feature_info = [dict(num_chs=inplanes, reduction=2, module='layer0')]
        self.layer1 = self._make_layer(
            block, planes=64, blocks=layers[0], groups=groups, reduction=reduction,
            downsample_kernel_size=1, downsample_padding=0)
        self.feature_info += [dict(num_chs=64 * block.expansion, reduction=4, module='layer1')]
        self.layer2 = self._make_layer(
            block, planes=128, blocks=layers[1], stride=2, groups=groups, reduction=reduction,
            downsample_kernel_size=downsample_kernel_size, downsample_padding=downsample_padding)
        self.feature_info += [dict(num_chs=128 * block.expansion, reduction=8, module='layer2')]
        self.layer3 = self._make_layer(
            block, planes=256, blocks=layers[2], stride=2, groups=groups, reduction=reduction,
            downsample_kernel_size=downsample_kernel_size, downsample_padding=downsample_padding)
        self.feature_info += [dict(num_chs=256 * block.expansion, reduction=16, module='layer3')]
        self.layer4 = self._make_layer(
            block, planes=512, blocks=layers[3], stride=2, groups=groups, reduction=reduction,
            downsample_kernel_size=downsample_kernel_size, downsample_padding=downsample_padding)
        self.feature_info += [dict(num_chs=512 * block.expansion, reduction=32, module='layer4')]
        self.num_features = self.head_hidden_size = 512 * block.expansion
        self.global_pool, self.last_linear = create_classifier(
            self.num_features, self.num_classes, pool_type=global_pool)

        for m in self.modules():
            _weight_init(m)

    def _make_layer(self, block, planes, blocks, groups, reduction, stride=1,
                    downsample_kernel_size=1, downsample_padding=0):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                nn.Conv2d(
                    self.inplanes, planes * block.expansion, kernel_size=downsample_kernel_size,
                    stride=stride, padding=downsample_padding, bias=False),
                nn.BatchNorm2d(planes * block.expansion),
            )

        layers = [block(self.inplanes, planes, groups, reduction, stride, downsample)]
        self.inplanes = planes * block.expansion
        for i in range(1, blocks):
            layers.append(block(self.inplanes, planes, groups, reduction))
        return nn.Sequential(*layers)

    @torch.jit.ignore
    def group_matcher(self, coarse=False):
        return dict(stem=r'^layer0', blocks=r'^layer(\d+)' if coarse else r'^layer(\d+)\.(\d+)')

    @torch.jit.ignore
    def set_grad_checkpointing(self, enable=True):
        assert not enable, 'gradient checkpointing not supported'

    @torch.jit.ignore
    def get_classifier(self) -> nn.Module:
        return self.last_linear

    def reset_classifier(self, num_classes: int, global_pool: str = 'avg'):
        self.num_classes = num_classes
        self.global_pool, self.last_linear = create_classifier(
            self.num_features, self.num_classes, pool_type=global_pool)

    def forward_features(self, x):
        x = self.layer0(x)
        x = self.pool0(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        return x

    def forward_head(self, x, pre_logits: bool = False):
        x = self.global_pool(x)
        if self.drop_rate > 0.:
            x = F.dropout(x, p=self.drop_rate, training=self.training)
        return x if pre_logits else self.last_linear(x)

    def forward(self, x):
        x = self.forward_features(x)
        x = self.forward_head(x)
        return x


def _create_senet(variant, pretrained=False, **kwargs):
    return build_model_with_cfg(SENet, variant, pretrained, **kwargs)


def _cfg(url='', **kwargs):
    return {
        'url': url, 'num_classes': 1000, 'input_size': (3, 224, 224), 'pool_size': (7, 7),
        'crop_pct': 0.875, 'interpolation': 'bilinear',
        'mean': IMAGENET_DEFAULT_MEAN, 'std': IMAGENET_DEFAULT_STD,
        'first_conv': 'layer0.conv1', 'classifier': 'last_linear',
        **kwargs,
    }


default_cfgs = generate_default_cfgs({
    'legacy_senet154.in1k': _cfg(),
    'legacy_seresnet18.in1k': _cfg(interpolation='bicubic'),
    'legacy_seresnet34.in1k': _cfg(),
    'legacy_seresnet50.in1k': _cfg(),
    'legacy_seresnet101.in1k': _cfg(),
    'legacy_seresnet152.in1k': _cfg(),
    'legacy_seresnext26_32x4d.in1k': _cfg(interpolation='bicubic'),
    'legacy_seresnext50_32x4d.in1k': _cfg(),
    'legacy_seresnext101_32x4d.in1k': _cfg(),
})


@register_model
def legacy_seresnet18(pretrained=False, **kwargs) -> SENet:
    model_args = dict(block=SEResNetBlock, layers=[2, 2, 2, 2], groups=1, reduction=16)
    return _create_senet('legacy_seresnet18', pretrained, **dict(model_args, **kwargs))


@register_model
def legacy_seresnet34(pretrained=False, **kwargs) -> SENet:
    model_args = dict(block=SEResNetBlock, layers=[3, 4, 6, 3], groups=1, reduction=16)
    return _create_senet('legacy_seresnet34', pretrained, **dict(model_args, **kwargs))


@register_model
def legacy_seresnet50(pretrained=False, **kwargs) -> SENet:
    model_args = dict(block=SEResNetBottleneck, layers=[3, 4, 6, 3], groups=1, reduction=16)
    return _create_senet('legacy_seresnet50', pretrained, **dict(model_args, **kwargs))


@register_model
def legacy_seresnet101(pretrained=False, **kwargs) -> SENet:
    model_args = dict(block=SEResNetBottleneck, layers=[3, 4, 23, 3], groups=1, reduction=16)
    return _create_senet('legacy_seresnet101', pretrained, **dict(model_args, **kwargs))


@register_model
def legacy_seresnet152(pretrained=False, **kwargs) -> SENet:
    model_args = dict(block=SEResNetBottleneck, layers=[3, 8, 36, 3], groups=1, reduction=16)
    return _create_senet('legacy_seresnet152', pretrained, **dict(model_args, **kwargs))


@register_model
def legacy_senet154(pretrained=False, **kwargs) -> SENet:
    model_args = dict(
        block=SEBottleneck, layers=[3, 8, 36, 3], groups=64, reduction=16,
        downsample_kernel_size=3, downsample_padding=1, inplanes=128, input_3x3=True)
    return _create_senet('legacy_senet154', pretrained, **dict(model_args, **kwargs))


@register_model
def legacy_seresnext26_32x4d(pretrained=False, **kwargs) -> SENet:
    model_args = dict(block=SEResNeXtBottleneck, layers=[2, 2, 2, 2], groups=32, reduction=16)
    return _create_senet('legacy_seresnext26_32x4d', pretrained, **dict(model_args, **kwargs))


@register_model
def legacy_seresnext50_32x4d(pretrained=False, **kwargs) -> SENet:
    model_args = dict(block=SEResNeXtBottleneck, layers=[3, 4, 6, 3], groups=32, reduction=16)
    return _create_senet('legacy_seresnext50_32x4d', pretrained, **dict(model_args, **kwargs))


@register_model
def legacy_seresnext101_32x4d(pretrained=False, **kwargs) -> SENet:
    model_args = dict(block=SEResNeXtBottleneck, layers=[3, 4, 23, 3], groups=32, reduction=16)
    return _create_senet('legacy_seresnext101_32x4d', pretrained, **dict(model_args, **kwargs))
