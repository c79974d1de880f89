"""NaFlex loader (reference `timm/data/naflex_loader.py`).

`NaFlexPrefetchLoader` (:27-222 — side HIP stream H2D + patch-layout-aware
normalize over P-P-C flattened patches) and `create_naflex_loader` (:225).
"""
import math
from contextlib import suppress
from functools import partial
from typing import Callable, Optional, Tuple, Union

import torch
import torch.utils.data

from .constants import IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD
from .naflex_dataset import NaFlexCollator, NaFlexMapDatasetWrapper
from .naflex_transforms import Patchify, RandomResizedCropToSequence, ResizeToSequence
from . import image_ops


class NaFlexPrefetchLoader:
    """Prefetcher for NaFlex batches: async H2D on a side HIP stream + on-GPU
    normalize aware of the P-P-C patch layout (reference `naflex_loader.py:27`)."""

    def __init__(
            self,
            loader,
            mean=IMAGENET_DEFAULT_MEAN,
            std=IMAGENET_DEFAULT_STD,
            img_dtype=torch.float32,
            device=torch.device('cuda'),
    ):
        self.loader = loader
        self.device = device
        self.img_dtype = img_dtype or torch.float32

        # patches are [B, N, P*P*C] with channel fastest (P-P-C layout)
        self.mean = torch.tensor(mean, device=device, dtype=self.img_dtype)
        self.std = torch.tensor(std, device=device, dtype=self.img_dtype)
        self.is_cuda = device.type == 'cuda' and torch.cuda.is_available()

    def _normalize(self, patches: torch.Tensor) -> torch.Tensor:
        C = self.mean.numel()
        shape = patches.shape
        p = patches.view(shape[0], shape[1], -1, C)
        p = p.sub_(self.mean).div_(self.std)
        return p.view(shape)

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
                next_input = {
                    k: (v.to(device=self.device, non_blocking=True) if isinstance(v, torch.Tensor) else v)
                    for k, v in next_input.items()
                }
                next_target = next_target.to(device=self.device, non_blocking=True)
                next_input['patches'] = self._normalize(next_input['patches'].to(self.img_dtype))

            if not first:
                yield input, target  # noqa: F821
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
        return getattr(self.loader, 'sampler', None)

    @property
    def dataset(self):
        return self.loader.dataset


def create_naflex_loader(
        dataset,
        patch_size: Union[Tuple[int, int], int] = 16,
        train_seq_lens: Tuple[int, ...] = (128, 256, 576, 784, 1024),
        max_seq_len: int = 576,
        mixup_fn: Optional[Callable] = None,
        batch_size: int = 32,  # used for eval & max for train
        max_tokens_per_batch: int = 4096 * 4,
        is_training: bool = False,
        mean=IMAGENET_DEFAULT_MEAN,
        std=IMAGENET_DEFAULT_STD,
        num_workers: int = 4,
        distributed: bool = False,
        rank: int = 0,
        world_size: int = 1,
        seed: int = 42,
        epoch: int = 0,
        pin_memory: bool = False,
        img_dtype: torch.dtype = torch.float32,
        device: torch.device = torch.device('cuda'),
        use_prefetcher: bool = True,
        persistent_workers: bool = True,
):
    """Create the NaFlex train/eval loader (reference `naflex_loader.py:225`)."""
    if isinstance(patch_size, (tuple, list)):
        patch_size = patch_size[0]

    if is_training:
        def transform_factory(max_seq_len, patch_size):
            def _t(img):
                img = RandomResizedCropToSequence(patch_size, max_seq_len)(img)
                img = image_ops.RandomHorizontalFlip()(img)
                return image_ops.to_tensor(img)
            return _t

        wrapped = NaFlexMapDatasetWrapper(
            dataset,
            patch_size=patch_size,
            seq_lens=train_seq_lens,
            max_tokens_per_batch=max_tokens_per_batch,
            transform_factory=transform_factory,
            mixup_fn=mixup_fn,
            seed=seed,
            distributed=distributed,
            rank=rank,
            world_size=world_size,
            epoch=epoch,
        )
        loader = torch.utils.data.DataLoader(
            wrapped,
            batch_size=None,  # batches are pre-collated by the wrapper
            num_workers=num_workers,
            pin_memory=pin_memory,
            persistent_workers=persistent_workers and num_workers > 0,
        )
    else:
        # fixed seq-len eval path: resize-to-seq + patchify per sample, pad-collate
        def eval_transform(img):
            img = ResizeToSequence(patch_size, max_seq_len)(img)
            t = image_ops.to_tensor(img)
            return Patchify(patch_size)(t)

        if hasattr(dataset, 'transform'):
            dataset.transform = eval_transform
        collator = NaFlexCollator(patch_size, max_seq_len=max_seq_len)
        sampler = None
        if distributed:
            from .distributed_sampler import OrderedDistributedSampler
            sampler = OrderedDistributedSampler(dataset)
        loader = torch.utils.data.DataLoader(
            dataset,
            batch_size=batch_size,
            shuffle=False,
            sampler=sampler,
            num_workers=num_workers,
            collate_fn=collator,
            pin_memory=pin_memory,
            persistent_workers=persistent_workers and num_workers > 0,
        )

    if use_prefetcher:
        loader = NaFlexPrefetchLoader(
            loader,
            mean=mean,
            std=std,
            img_dtype=img_dtype,
            device=device,
        )
    return loader
