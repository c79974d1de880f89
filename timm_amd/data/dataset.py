"""Dataset wrappers bridging sample readers to torch Datasets.

Behavioral parity: /root/reference/timm/data/dataset.py:21,90,170
(error-retry decode loop, reader pass-through config, AugMix split tuples).
"""
import logging
from typing import Optional

import torch
import torch.utils.data as data
from PIL import Image

from .readers import create_reader

_logger = logging.getLogger(__name__)

# give up after this many consecutive undecodable samples
_ERROR_RETRY = 50

__all__ = ['ImageDataset', 'IterableImageDataset', 'AugMixDataset']


class ImageDataset(data.Dataset):
    """Map-style dataset: indexes a reader, decodes, applies transforms.

    Undecodable samples are skipped (advancing to the next index) up to
    _ERROR_RETRY consecutive failures.
    """

    def __init__(
            self,
            root,
            reader=None,
            split='train',
            class_map=None,
            load_bytes=False,
            input_img_mode='RGB',
            transform=None,
            target_transform=None,
            **kwargs,
    ):
        if reader is None or isinstance(reader, str):
            reader = create_reader(
                reader or '', root=root, split=split, class_map=class_map, **kwargs)
        self.reader = reader
        self.load_bytes = load_bytes
        self.input_img_mode = input_img_mode
        self.transform = transform
        self.target_transform = target_transform
        self._consecutive_errors = 0

    def _decode(self, index, img_io):
        try:
            return img_io.read() if self.load_bytes else Image.open(img_io)
        except Exception as e:
            _logger.warning(
                f'Skipped sample (index {index}, file {self.reader.filename(index)}). {str(e)}')
            self._consecutive_errors += 1
            if self._consecutive_errors >= _ERROR_RETRY:
                raise e
            return None

    def __getitem__(self, index):
        img_io, target = self.reader[index]
        img = self._decode(index, img_io)
        if img is None:
            return self.__getitem__((index + 1) % len(self.reader))
        self._consecutive_errors = 0

        if self.input_img_mode and not self.load_bytes:
            img = img.convert(self.input_img_mode)
        if self.transform is not None:
            img = self.transform(img)
        if target is None:
            target = -1
        elif self.target_transform is not None:
            target = self.target_transform(target)
        return img, target

    def __len__(self):
        return len(self.reader)

    def filename(self, index, basename=False, absolute=False):
        return self.reader.filename(index, basename, absolute)

    def filenames(self, basename=False, absolute=False):
        return self.reader.filenames(basename, absolute)


class IterableImageDataset(data.IterableDataset):
    """Stream-style dataset over shardable readers (tar/wds/tfds/hfds)."""

    # reader kwargs forwarded verbatim when built from a spec string
    _READER_KEYS = (
        'is_training', 'batch_size', 'num_samples', 'seed', 'repeats',
        'download', 'input_img_mode', 'input_key', 'target_key', 'max_steps',
    )

    def __init__(
            self,
            root,
            reader=None,
            split='train',
            class_map=None,
            is_training=False,
            batch_size=1,
            num_samples=None,
            seed=42,
            repeats=0,
            download=False,
            input_img_mode='RGB',
            input_key=None,
            target_key=None,
            transform=None,
            target_transform=None,
            max_steps=None,
    ):
        assert reader is not None
        if isinstance(reader, str):
            local = locals()
            reader = create_reader(
                reader, root=root, split=split, class_map=class_map,
                **{k: local[k] for k in self._READER_KEYS})
        self.reader = reader
        self.transform = transform
        self.target_transform = target_transform
        self._consecutive_errors = 0

    def __iter__(self):
        for img, target in self.reader:
            if self.transform is not None:
                img = self.transform(img)
            if self.target_transform is not None:
                target = self.target_transform(target)
            yield img, target

    def __len__(self):
        return len(self.reader) if hasattr(self.reader, '__len__') else 0

    def set_epoch(self, count):
        """Propagate epoch for deterministic cross-worker shuffle (tfds/wds)."""
        if hasattr(self.reader, 'set_epoch'):
            self.reader.set_epoch(count)

    def set_loader_cfg(self, num_workers: Optional[int] = None):
        """Propagate loader geometry for per-worker sample-count estimates."""
        if hasattr(self.reader, 'set_loader_cfg'):
            self.reader.set_loader_cfg(num_workers=num_workers)

    def filename(self, index, basename=False, absolute=False):
        raise AssertionError('Filename lookup by index not supported, use filenames().')

    def filenames(self, basename=False, absolute=False):
        return self.reader.filenames(basename, absolute)


class AugMixDataset(torch.utils.data.Dataset):
    """Wraps a dataset to emit (clean, aug1, ..., augN-1) tuples for JSD-style
    consistency training.  ``transform`` must be set to a 3-tuple of
    (base, augmentation, normalize) transforms."""

    def __init__(self, dataset, num_splits=2):
        self.dataset = dataset
        self.num_splits = num_splits
        self.augmentation = None
        self.normalize = None
        if dataset.transform is not None:
            self._set_transforms(dataset.transform)

    def _set_transforms(self, t):
        assert isinstance(t, (list, tuple)) and len(t) == 3, \
            'Expecting a tuple/list of 3 transforms'
        self.dataset.transform, self.augmentation, self.normalize = t

    @property
    def transform(self):
        return self.dataset.transform

    @transform.setter
    def transform(self, t):
        self._set_transforms(t)

    def _finalize(self, x):
        return x if self.normalize is None else self.normalize(x)

    def __getitem__(self, i):
        x, y = self.dataset[i]  # base transform applied by the inner dataset
        splits = [self._finalize(x)]  # clean split: normalize only
        splits += [
            self._finalize(self.augmentation(x)) for _ in range(self.num_splits - 1)]
        return tuple(splits), y

    def __len__(self):
        return len(self.dataset)
