"""Device-side input pipeline (reference `timm/data/loader.py`).

`fast_collate` (:30) assembles uint8 batches; `PrefetchLoader` (:81-159) is
the MI355X-native prefetcher: a dedicated side HIP stream does the async H2D
copy + on-GPU dequant-normalize `(x-mean)/std` + RandomErasing, with a
stream-wait handshake so compute consumes tensors without a sync.
`create_loader` (:205) wires samplers/transforms/collate; `MultiEpochsDataLoader`
(:472) reuses workers across epochs.
"""
import logging
import random
from contextlib import suppress
from functools import partial
from itertools import repeat
from typing import Any, Callable, Iterator, List, Optional, Sequence, Tuple, Union

import torch

from .. import ops
import torch.utils.data
import numpy as np

from .constants import IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD
from .dataset import IterableImageDataset, ImageDataset
from .distributed_sampler import OrderedDistributedSampler, RepeatAugSampler
from .scheduled_sampler import ScheduledBatchSampler, ScheduledTransformDataset
from .random_erasing import RandomErasing
from .mixup import FastCollateMixup
from .transforms_factory import create_transform

_logger = logging.getLogger(__name__)


def fast_collate(batch):
    """uint8 batch assembly without the default-collate overhead
    (reference `loader.py:30`). Handles numpy arrays, torch tensors, and
    aug-split tuples of arrays (flattened split-major so torch.split by
    batch_size recovers each split)."""
    assert isinstance(batch[0], tuple)
    batch_size = len(batch)
    sample0 = batch[0][0]

    if isinstance(sample0, tuple):
        # aug-split samples: position j of every sample lands in slot
        # [j * batch_size + i] so splits stay contiguous
        inner = len(sample0)
        targets = torch.zeros(batch_size * inner, dtype=torch.int64)
        tensor = torch.zeros((batch_size * inner, *sample0[0].shape), dtype=torch.uint8)
        for i, (imgs, target) in enumerate(batch):
            assert len(imgs) == inner
            for j, img in enumerate(imgs):
                targets[i + j * batch_size] = target
                tensor[i + j * batch_size] += torch.from_numpy(img)
        return tensor, targets

    targets = torch.tensor([b[1] for b in batch], dtype=torch.int64)
    tensor = torch.zeros((batch_size, *sample0.shape), dtype=torch.uint8)
    if isinstance(sample0, np.ndarray):
        for i, (img, _) in enumerate(batch):
            tensor[i] += torch.from_numpy(img)
    elif isinstance(sample0, torch.Tensor):
        for i, (img, _) in enumerate(batch):
            tensor[i].copy_(img)
    else:
        raise AssertionError(f'Unexpected sample type {type(sample0)}')
    return tensor, targets


def adapt_to_chs(x, n):
    """Broadcast / reconcile per-channel normalization stats to n channels."""
    if not isinstance(x, (tuple, list)):
        return tuple(repeat(x, n))
    if len(x) != n:
        mean_val = np.mean(x).item()
        _logger.warning(f'Pretrained mean/std different shape than model, using avg value {(mean_val,) * n}.')
        return (mean_val,) * n
    return x


class PrefetchLoader:
    """Prefetcher with a dedicated side HIP stream: async H2D + fused
    dequant-normalize + RandomErasing overlapped with compute
    (reference `loader.py:81-159`)."""

    def __init__(
            self,
            loader: torch.utils.data.DataLoader,
            mean: Tuple[float, ...] = IMAGENET_DEFAULT_MEAN,
            std: Tuple[float, ...] = IMAGENET_DEFAULT_STD,
            channels: int = 3,
            device: torch.device = torch.device('cuda'),
            img_dtype: Optional[torch.dtype] = None,
            fp16: bool = False,
            re_prob: float = 0.,
            re_mode: str = 'const',
            re_count: int = 1,
            re_num_splits: int = 0,
    ):
        mean = adapt_to_chs(mean, channels)
        std = adapt_to_chs(std, channels)
        normalization_shape = (1, channels, 1, 1)

        self.loader = loader
        self.device = device
        if fp16:
            # fp16 arg is deprecated, but will override dtype arg if set for bwd compat
            img_dtype = torch.float16
        self.img_dtype = img_dtype or torch.float32
        self.mean = torch.tensor(
            [x * 255 for x in mean], device=device, dtype=self.img_dtype).view(normalization_shape)
        self.std = torch.tensor(
            [x * 255 for x in std], device=device, dtype=self.img_dtype).view(normalization_shape)
        if re_prob > 0.:
            self.random_erasing = RandomErasing(
                probability=re_prob,
                mode=re_mode,
                max_count=re_count,
                num_splits=re_num_splits,
                device=device,
            )
        else:
            self.random_erasing = None
        self.is_cuda = device.type == 'cuda' and torch.cuda.is_available()

    def __iter__(self):
        first = True
        if self.is_cuda:
            stream = torch.cuda.Stream()
            stream_context = partial(torch.cuda.stream, stream=stream)
        else:
            stream = None
            stream_context = suppress

        for next_input, next_target in self.loader:
            with stream_context():
                next_input = next_input.to(device=self.device, non_blocking=True)
                next_target = next_target.to(device=self.device, non_blocking=True)
                if next_input.dtype == torch.uint8 and next_input.dim() == 4:
                    # fused uint8 -> normalized tensor (one HIP kernel, csrc/data_ops.hip)
                    next_input = ops.u8_normalize(next_input, self.mean, self.std, self.img_dtype)
                else:
                    next_input = next_input.to(self.img_dtype).sub_(self.mean).div_(self.std)
                if self.random_erasing is not None:
                    next_input = self.random_erasing(next_input)

            if not first:
                yield input, target  # noqa: F821, F823
            else:
                first = False

            if stream is not None:
                torch.cuda.current_stream().wait_stream(stream)

            input = next_input
            target = next_target

        yield input, target

    def __len__(self):
        return len(self.loader)

    @property
    def sampler(self):
        return self.loader.sampler

    @property
    def dataset(self):
        return self.loader.dataset

    @property
    def mixup_enabled(self):
        if isinstance(self.loader.collate_fn, FastCollateMixup):
            return self.loader.collate_fn.mixup_enabled
        else:
            return False

    @mixup_enabled.setter
    def mixup_enabled(self, x):
        if isinstance(self.loader.collate_fn, FastCollateMixup):
            self.loader.collate_fn.mixup_enabled = x


def _worker_init(worker_id, worker_seeding='all'):
    worker_info = torch.utils.data.get_worker_info()
    assert worker_info.id == worker_id
    if isinstance(worker_seeding, Callable):
        seed = worker_seeding(worker_info)
        random.seed(seed)
        torch.manual_seed(seed)
        np.random.seed(seed % (2 ** 32 - 1))
    else:
        assert worker_seeding in ('all', 'part')
        # random / torch seed already called in dataloader iter class w/ worker_info.seed
        # to reproduce some old results (same seed + hparam combo), partial seeding is required (skip numpy re-seed)
        if worker_seeding == 'all':
            np.random.seed(worker_info.seed % (2 ** 32 - 1))


def create_loader(
        dataset: Union[ImageDataset, IterableImageDataset],
        input_size: Union[int, Tuple[int, int], Tuple[int, int, int]],
        batch_size: int,
        is_training: bool = False,
        no_aug: bool = False,
        re_prob: float = 0.,
        re_mode: str = 'const',
        re_count: int = 1,
        re_split: bool = False,
        train_crop_mode: Optional[str] = None,
        scale: Optional[Tuple[float, float]] = None,
        ratio: Optional[Tuple[float, float]] = None,
        hflip: float = 0.5,
        vflip: float = 0.,
        color_jitter: float = 0.4,
        color_jitter_prob: Optional[float] = None,
        grayscale_prob: float = 0.,
        gaussian_blur_prob: float = 0.,
        auto_augment: Optional[str] = None,
        num_aug_repeats: int = 0,
        num_aug_splits: int = 0,
        interpolation: str = 'bilinear',
        mean: Tuple[float, ...] = IMAGENET_DEFAULT_MEAN,
        std: Tuple[float, ...] = IMAGENET_DEFAULT_STD,
        num_workers: int = 1,
        distributed: bool = False,
        crop_pct: Optional[float] = None,
        crop_mode: Optional[str] = None,
        crop_border_pixels: Optional[int] = None,
        collate_fn: Optional[Callable] = None,
        pin_memory: bool = False,
        fp16: bool = False,  # deprecated, use img_dtype
        img_dtype: torch.dtype = torch.float32,
        device: torch.device = torch.device('cuda'),
        use_prefetcher: bool = True,
        use_multi_epochs_loader: bool = False,
        persistent_workers: bool = True,
        worker_seeding: str = 'all',
        tf_preprocessing: bool = False,
        input_size_choices: Optional[Sequence] = None,
        batch_size_choices: Optional[Sequence[int]] = None,
        batch_choice_weights: Optional[Sequence[float]] = None,
        batch_choice_seed: int = 0,
        batch_choice_schedule: str = 'constant',
        batch_schedule_epochs: Optional[int] = None,
        batch_schedule_spread: float = 0.65,
        batch_schedule_random_mix: float = 0.1,
        num_batches: Optional[int] = None,
):
    """Create the train/eval loader (reference `loader.py:205-469`)."""
    re_num_splits = 0
    if re_split:
        # apply RE to second half of batch if no aug split otherwise line up with aug split
        re_num_splits = num_aug_splits or 2

    if isinstance(dataset, IterableImageDataset):
        # give Iterable datasets early knowledge of num_workers so that sample estimates
        # are correct before worker processes are launched
        dataset.set_loader_cfg(num_workers=num_workers)

    scheduled_batching = input_size_choices is not None
    if scheduled_batching:
        if not is_training:
            raise ValueError('Scheduled input sizes are only supported for training loaders.')
        if use_multi_epochs_loader:
            raise ValueError('MultiEpochsDataLoader is not supported with scheduled input sizes.')
        if isinstance(dataset, torch.utils.data.IterableDataset):
            raise TypeError('Scheduled input sizes require a map-style dataset.')
        if num_aug_splits > 0:
            raise ValueError('Augmentation splits are not supported with scheduled input sizes.')
        if not input_size_choices:
            raise ValueError('input_size_choices must contain at least one size.')
        channels = input_size[0] if isinstance(input_size, (tuple, list)) and len(input_size) == 3 else len(mean)
        resolved_input_sizes = []
        for size in input_size_choices:
            if isinstance(size, int):
                size = (channels, size, size)
            elif len(size) == 2:
                size = (channels, *size)
            elif len(size) == 3:
                size = tuple(size)
                if size[0] != channels:
                    raise ValueError('All scheduled input sizes must use the same number of channels.')
            else:
                raise ValueError('Scheduled input sizes must be scalars, HW tuples, or CHW tuples.')
            if any(d <= 0 for d in size):
                raise ValueError('All scheduled input size dimensions must be positive.')
            resolved_input_sizes.append(size)
        if batch_size_choices is None:
            batch_size_choices = [batch_size] * len(resolved_input_sizes)
        elif len(batch_size_choices) != len(resolved_input_sizes):
            raise ValueError('batch_size_choices and input_size_choices must have the same length.')
        if batch_choice_weights is not None and len(batch_choice_weights) != len(resolved_input_sizes):
            raise ValueError('batch_choice_weights and input_size_choices must have the same length.')

    if getattr(dataset, 'transform', None) is None or not tf_preprocessing:
        def _make_transform(_size):
            return create_transform(
            _size,
            is_training=is_training,
            no_aug=no_aug,
            train_crop_mode=train_crop_mode,
            scale=scale,
            ratio=ratio,
            hflip=hflip,
            vflip=vflip,
            color_jitter=color_jitter,
            color_jitter_prob=color_jitter_prob,
            grayscale_prob=grayscale_prob,
            gaussian_blur_prob=gaussian_blur_prob,
            auto_augment=auto_augment,
            interpolation=interpolation,
            mean=mean,
            std=std,
            crop_pct=crop_pct,
            crop_mode=crop_mode,
            crop_border_pixels=crop_border_pixels,
            re_prob=re_prob,
            re_mode=re_mode,
            re_count=re_count,
            re_num_splits=re_num_splits,
            tf_preprocessing=tf_preprocessing,
            use_prefetcher=use_prefetcher,
            separate=num_aug_splits > 0,
        )

        if scheduled_batching:
            transforms = [_make_transform(size) for size in resolved_input_sizes]
            dataset.transform = None
            dataset = ScheduledTransformDataset(dataset, transforms)
        else:
            dataset.transform = _make_transform(input_size)

    if isinstance(dataset, IterableImageDataset):
        # wrap dataset in AugMix helper
        if num_aug_splits > 1:
            raise NotImplementedError('AugMix not currently supported for iterable datasets')

    sampler = None
    if distributed and not isinstance(dataset, torch.utils.data.IterableDataset):
        if is_training:
            if num_aug_repeats:
                sampler = RepeatAugSampler(dataset, num_repeats=num_aug_repeats)
            else:
                sampler = torch.utils.data.distributed.DistributedSampler(dataset)
        else:
            # This will add extra duplicate entries to result in equal num
            # of samples per-process, will slightly alter validation results
            sampler = OrderedDistributedSampler(dataset)
    else:
        assert num_aug_repeats == 0, "RepeatAugment not currently supported in non-distributed or IterableDataset use"

    if scheduled_batching and sampler is None:
        sampler = torch.utils.data.RandomSampler(dataset)

    if collate_fn is None:
        collate_fn = fast_collate if use_prefetcher else torch.utils.data.dataloader.default_collate

    loader_class = torch.utils.data.DataLoader
    if use_multi_epochs_loader:
        loader_class = MultiEpochsDataLoader

    if scheduled_batching:
        loader_args = dict(
            batch_sampler=ScheduledBatchSampler(
                sampler,
                batch_sizes=batch_size_choices,
                choice_weights=batch_choice_weights,
                seed=batch_choice_seed,
                drop_last=is_training,
                num_batches=num_batches,
                choice_schedule=batch_choice_schedule,
                schedule_epochs=batch_schedule_epochs,
                schedule_spread=batch_schedule_spread,
                schedule_random_mix=batch_schedule_random_mix,
            ),
            num_workers=num_workers,
            collate_fn=collate_fn,
            pin_memory=pin_memory,
            worker_init_fn=partial(_worker_init, worker_seeding=worker_seeding),
            persistent_workers=persistent_workers and num_workers > 0,
        )
    else:
        loader_args = dict(
            batch_size=batch_size,
            shuffle=not isinstance(dataset, torch.utils.data.IterableDataset) and sampler is None and is_training,
            num_workers=num_workers,
            sampler=sampler,
            collate_fn=collate_fn,
            pin_memory=pin_memory,
            drop_last=is_training,
            worker_init_fn=partial(_worker_init, worker_seeding=worker_seeding),
            persistent_workers=persistent_workers and num_workers > 0,
        )
    try:
        loader = loader_class(dataset, **loader_args)
    except TypeError:
        loader_args.pop('persistent_workers')  # only in Pytorch 1.7+
        loader = loader_class(dataset, **loader_args)

    if use_prefetcher:
        prefetch_re_prob = re_prob if is_training and not no_aug else 0.
        loader = PrefetchLoader(
            loader,
            mean=mean,
            std=std,
            channels=input_size[0] if isinstance(input_size, (tuple, list)) else 3,
            device=device,
            fp16=fp16,  # deprecated, use img_dtype
            img_dtype=img_dtype,
            re_prob=prefetch_re_prob,
            re_mode=re_mode,
            re_count=re_count,
            re_num_splits=re_num_splits,
        )

    return loader


class MultiEpochsDataLoader(torch.utils.data.DataLoader):
    """DataLoader that keeps workers alive across epochs (reference `loader.py:472`)."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self._DataLoader__initialized = False
        if self.batch_sampler is None:
            self.sampler = _RepeatSampler(self.sampler)
        else:
            self.batch_sampler = _RepeatSampler(self.batch_sampler)
        self._DataLoader__initialized = True
        self.iterator = super().__iter__()

    def __len__(self):
        return len(self.sampler) if self.batch_sampler is None else len(self.batch_sampler.sampler)

    def __iter__(self):
        for i in range(len(self)):
            yield next(self.iterator)


class _RepeatSampler(object):
    """Sampler that repeats forever."""

    def __init__(self, sampler):
        self.sampler = sampler

    def __iter__(self):
        while True:
            yield from iter(self.sampler)
