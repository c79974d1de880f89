"""Import the reference timm (/root/reference) for behavior-parity testing.

The reference imports torchvision at package level; this environment has no
torchvision, so minimal stubs are installed first.  Test-only helper — the
framework itself never imports the reference.
"""
import sys
import types

import torch


def _stub_torchvision():
    if 'torchvision' in sys.modules:
        return
    tv = types.ModuleType('torchvision')
    ops = types.ModuleType('torchvision.ops')
    misc = types.ModuleType('torchvision.ops.misc')

    class FrozenBatchNorm2d(torch.nn.Module):
        def __init__(self, num_features, eps=1e-5):
            super().__init__()
            self.eps = eps
            self.register_buffer('weight', torch.ones(num_features))
            self.register_buffer('bias', torch.zeros(num_features))
            self.register_buffer('running_mean', torch.zeros(num_features))
            self.register_buffer('running_var', torch.ones(num_features))

        def forward(self, x):
            scale = self.weight * (self.running_var + self.eps).rsqrt()
            return x * scale.reshape(1, -1, 1, 1) + \
                (self.bias - self.running_mean * scale).reshape(1, -1, 1, 1)

    misc.FrozenBatchNorm2d = FrozenBatchNorm2d
    ops.misc = misc
    tv.ops = ops

    transforms = types.ModuleType('torchvision.transforms')
    functional = types.ModuleType('torchvision.transforms.functional')

    class InterpolationMode:
        NEAREST = 'nearest'
        BILINEAR = 'bilinear'
        BICUBIC = 'bicubic'
        BOX = 'box'
        HAMMING = 'hamming'
        LANCZOS = 'lanczos'

    functional.InterpolationMode = InterpolationMode
    transforms.functional = functional
    transforms.InterpolationMode = InterpolationMode

    class _Noop:
        def __init__(self, *a, **k):
            pass

        def __call__(self, x):
            return x

    for name in ('Compose', 'ToTensor', 'Normalize', 'Resize', 'CenterCrop',
                 'RandomCrop', 'RandomHorizontalFlip', 'RandomVerticalFlip',
                 'ColorJitter', 'RandomResizedCrop', 'PILToTensor'):
        setattr(transforms, name, type(name, (_Noop,), {}))
    tv.transforms = transforms

    models = types.ModuleType('torchvision.models')
    tv.models = models

    datasets = types.ModuleType('torchvision.datasets')

    class _StubDataset:
        def __init__(self, *a, **k):
            raise RuntimeError('torchvision stub: datasets unavailable')

    for name in ('CIFAR100', 'CIFAR10', 'MNIST', 'KMNIST', 'FashionMNIST',
                 'ImageFolder', 'ImageNet', 'Places365', 'QMNIST'):
        setattr(datasets, name, type(name, (_StubDataset,), {}))
    tv.datasets = datasets
    tv.__version__ = '0.0.0-stub'
    tv.__path__ = []  # mark as package so "from torchvision.datasets import X" resolves

    sys.modules['torchvision'] = tv
    sys.modules['torchvision.ops'] = ops
    sys.modules['torchvision.ops.misc'] = misc
    sys.modules['torchvision.transforms'] = transforms
    sys.modules['torchvision.transforms.functional'] = functional
    sys.modules['torchvision.models'] = models
    sys.modules['torchvision.datasets'] = datasets


def load_reference():
    """Returns the reference `timm` package (imported from /root/reference)."""
    _stub_torchvision()
    if '/root/reference' not in sys.path:
        sys.path.insert(0, '/root/reference')
    import timm  # noqa
    return timm
