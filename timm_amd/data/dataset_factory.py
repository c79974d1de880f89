"""Dataset factory: dispatch ``<type>/<name>`` specs to dataset backends.

Behavioral parity: /root/reference/timm/data/dataset_factory.py:63
(same spec grammar and split-name normalization).  Structure here is
table-driven: torchvision builtins and streaming reader prefixes each map
through small builder helpers instead of one long if/elif chain.
"""
import os
from typing import Optional

from .dataset import ImageDataset, IterableImageDataset

try:
    import torchvision.datasets as tvd
    has_torchvision = True
except ImportError:
    tvd = None
    has_torchvision = False

_TRAIN_SYNONYM = ('train', 'training')
_EVAL_SYNONYM = ('val', 'valid', 'validation', 'eval', 'evaluation')

# torchvision datasets taking a train=bool flag
_TV_TRAIN_FLAG = ('cifar10', 'cifar100', 'mnist', 'kmnist', 'fashion_mnist', 'qmnist')
_TV_CLASSES = dict(
    cifar10='CIFAR10', cifar100='CIFAR100', mnist='MNIST', kmnist='KMNIST',
    fashion_mnist='FashionMNIST', qmnist='QMNIST', places365='Places365',
    imagenet='ImageNet', inaturalist='INaturalist', inat='INaturalist',
    image_folder='ImageFolder', folder='ImageFolder',
)


def _search_split(root, split):
    """Prefer a split-named (or synonym-named) subdir of root when present."""
    base = split.split('[')[0]
    direct = os.path.join(root, base)
    if os.path.exists(direct):
        return direct
    synonyms = ()
    if base in _TRAIN_SYNONYM:
        synonyms = _TRAIN_SYNONYM
    elif base in _EVAL_SYNONYM:
        synonyms = _EVAL_SYNONYM
    for name in synonyms:
        candidate = os.path.join(root, name)
        if os.path.exists(candidate):
            return candidate
    return root


def _tv_class(name):
    assert has_torchvision, 'torchvision is required for torch/* dataset names'
    cls = getattr(tvd, _TV_CLASSES.get(name, ''), None)
    assert cls is not None, (
        f'Unknown torchvision dataset {name}' if name not in _TV_CLASSES else
        f'Please update torchvision for the {name} dataset.')
    return cls


def _build_torchvision(name, split, root, download, search_split, kwargs):
    common = dict(root=root, download=download, **kwargs)
    if name in _TV_TRAIN_FLAG:
        return _tv_class(name)(train=split in _TRAIN_SYNONYM, **common)
    if name in ('inaturalist', 'inat'):
        # split may carry a target-type prefix: '<type>/<version>'
        target_type = 'full'
        parts = split.split('/')
        if len(parts) > 1:
            target_type = parts[0].split('_')
            if len(target_type) == 1:
                target_type = target_type[0]
            split = parts[-1]
        if split in _TRAIN_SYNONYM:
            split = '2021_train'
        elif split in _EVAL_SYNONYM:
            split = '2021_valid'
        return _tv_class(name)(version=split, target_type=target_type, **common)
    if name == 'places365':
        if split in _TRAIN_SYNONYM:
            split = 'train-standard'
        elif split in _EVAL_SYNONYM:
            split = 'val'
        return _tv_class(name)(split=split, **common)
    if name == 'imagenet':
        if split in _EVAL_SYNONYM:
            split = 'val'
        return _tv_class(name)(split=split, **common)
    if name in ('image_folder', 'folder'):
        if search_split and os.path.isdir(root):
            root = _search_split(root, split)
        return _tv_class(name)(root, **kwargs)
    return _tv_class(name)  # raises 'Unknown torchvision dataset'


def create_dataset(
        name: str,
        root: Optional[str] = None,
        split: str = 'validation',
        search_split: bool = True,
        class_map: dict = None,
        load_bytes: bool = False,
        is_training: bool = False,
        download: bool = False,
        batch_size: int = 1,
        num_samples: Optional[int] = None,
        seed: int = 42,
        repeats: int = 0,
        input_img_mode: str = 'RGB',
        **kwargs,
):
    """Build a dataset from a ``<type>/<name>`` spec.

    Types: ``torch/*`` torchvision builtins · ``hfds/*`` HF arrow (map-style)
    · ``hfids/*`` / ``tfds/*`` / ``wds/*`` streaming readers · anything else =
    folder/tar path handled by ImageDataset (with split-subdir search).
    """
    kwargs = {k: v for k, v in kwargs.items() if v is not None}
    name = (name or '').lower()

    if name.startswith('torch/'):
        return _build_torchvision(
            name.split('/', 2)[-1], split, root, download, search_split, kwargs)

    stream_args = dict(
        split=split,
        class_map=class_map,
        is_training=is_training,
        batch_size=batch_size,
        num_samples=num_samples,
        repeats=repeats,
        seed=seed,
        input_img_mode=input_img_mode,
    )
    if name.startswith('hfds/'):
        # HF arrow datasets are random-access; the reader handles decode
        return ImageDataset(
            root, reader=name, split=split, class_map=class_map,
            input_img_mode=input_img_mode, **kwargs)
    if name.startswith('hfids/') or name.startswith('wds/'):
        return IterableImageDataset(root, reader=name, **stream_args, **kwargs)
    if name.startswith('tfds/'):
        return IterableImageDataset(
            root, reader=name, download=download, **stream_args, **kwargs)

    # plain folder / tar path
    if search_split and os.path.isdir(root):
        root = _search_split(root, split)
    return ImageDataset(
        root, reader=name, class_map=class_map, load_bytes=load_bytes,
        input_img_mode=input_img_mode, **kwargs)
