"""NaFlex dataset wrapper + collator (reference `timm/data/naflex_dataset.py`).

`calculate_naflex_batch_size` (:31 — batch ∝ token budget / seq_len),
`NaFlexCollator` (:74 — pads patch seqs to the bucket seq-len, builds
patch_coord/patch_valid), `NaFlexMapDatasetWrapper` (:200 — IterableDataset
emitting *pre-collated batches* on a canonical distributed-aware schedule of
(seq_len, batch) pairs).
"""
import math
import random
import warnings
from typing import Any, Callable, Dict, Iterator, List, Optional, Tuple, Union

import torch
from torch.utils.data import Dataset, IterableDataset, get_worker_info

from .naflex_transforms import Patchify, RandomResizedCropToSequence, ResizeToSequence


def calculate_naflex_batch_size(
        tokens_per_batch: int,
        seq_len: int,
        max_size: Optional[int] = None,
        divisor: int = 1,
        rounding: str = 'floor',
) -> int:
    """Batch size for a seq-len bucket from a token budget (reference `:31`)."""
    batch_size = tokens_per_batch / seq_len
    if rounding == 'floor':
        batch_size = math.floor(batch_size / divisor) * divisor
    elif rounding == 'ceil':
        batch_size = math.ceil(batch_size / divisor) * divisor
    else:
        batch_size = round(batch_size / divisor) * divisor
    batch_size = max(batch_size, divisor)
    if max_size is not None:
        batch_size = min(batch_size, max_size)
    return int(batch_size)


class NaFlexCollator:
    """Pads a list of {patches, patch_coord, patch_valid} samples to a common
    seq-len and stacks (reference `:74-155`)."""

    def __init__(self, patch_size: int = 16, max_seq_len: Optional[int] = None):
        self.patch_size = patch_size
        self.max_seq_len = max_seq_len

    def __call__(self, batch) -> Tuple[Dict[str, torch.Tensor], torch.Tensor]:
        assert isinstance(batch[0], tuple)
        samples = [b[0] for b in batch]
        if isinstance(batch[0][1], torch.Tensor) and batch[0][1].ndim > 0:
            # soft targets from NaFlexMixup
            targets = torch.stack([b[1] for b in batch])
        else:
            targets = torch.tensor([b[1] for b in batch], dtype=torch.int64)

        seq_len = self.max_seq_len or max(s['patches'].shape[0] for s in samples)
        patch_dim = samples[0]['patches'].shape[-1]
        B = len(samples)
        patches = torch.zeros(B, seq_len, patch_dim, dtype=samples[0]['patches'].dtype)
        patch_coord = torch.zeros(B, seq_len, 2, dtype=torch.long)
        patch_valid = torch.zeros(B, seq_len, dtype=torch.bool)
        for i, s in enumerate(samples):
            n = min(s['patches'].shape[0], seq_len)
            patches[i, :n] = s['patches'][:n]
            patch_coord[i, :n] = s['patch_coord'][:n]
            patch_valid[i, :n] = s['patch_valid'][:n]
        return {
            'patches': patches,
            'patch_coord': patch_coord,
            'patch_valid': patch_valid,
            'seq_len': seq_len,
        }, targets


class NaFlexMapDatasetWrapper(IterableDataset):
    """Wraps a map-style dataset to emit pre-collated variable-seq-len batches
    (reference `:200`).

    Epoch schedule: shuffled list of (seq_len, batch_indices) assignments,
    deterministic per (seed, epoch), sharded across distributed ranks and
    dataloader workers.  Per-rank batch sizes vary with the seq-len bucket
    (constant token budget) — the train loop rescales the loss accordingly.
    """

    def __init__(
            self,
            base_dataset: Dataset,
            patch_size: int = 16,
            seq_lens: Tuple[int, ...] = (128, 256, 576, 784, 1024),
            max_tokens_per_batch: int = 4096 * 4,
            transform_factory: Optional[Callable] = None,
            mixup_fn: Optional[Callable] = None,
            seed: int = 42,
            shuffle: bool = True,
            distributed: bool = False,
            rank: int = 0,
            world_size: int = 1,
            epoch: int = 0,
            batch_divisor: int = 8,
    ):
        super().__init__()
        self.base_dataset = base_dataset
        self.patch_size = patch_size
        self.seq_lens = tuple(seq_lens)
        self.max_tokens_per_batch = max_tokens_per_batch
        self.seed = seed
        self.shuffle = shuffle
        self.distributed = distributed
        self.rank = rank
        self.world_size = world_size
        self.epoch = epoch
        self.batch_divisor = batch_divisor
        self.mixup_fn = mixup_fn

        if transform_factory is not None:
            self.transforms = {
                seq_len: transform_factory(max_seq_len=seq_len, patch_size=patch_size)
                for seq_len in self.seq_lens
            }
        else:
            self.transforms = {
                seq_len: None for seq_len in self.seq_lens
            }
        self.patchifier = Patchify(patch_size)
        self.collators = {
            seq_len: NaFlexCollator(patch_size, max_seq_len=seq_len) for seq_len in self.seq_lens
        }

        # canonical batch schedule, recomputed per epoch (reference `:307-486`)
        self._batches: List[Tuple[int, List[int]]] = []
        self._num_batches = 0
        self._update_schedule()

    def _update_schedule(self):
        g = torch.Generator()
        g.manual_seed(self.seed + self.epoch)
        n = len(self.base_dataset)
        indices = torch.randperm(n, generator=g).tolist() if self.shuffle else list(range(n))

        # assign a seq-len bucket to each chunk of samples, batch size from the token budget
        batches = []
        pos = 0
        while pos < n:
            seq_len = self.seq_lens[int(torch.randint(len(self.seq_lens), (1,), generator=g))]
            bs = calculate_naflex_batch_size(
                self.max_tokens_per_batch, seq_len, divisor=self.batch_divisor)
            # global batch split across ranks
            bs_global = bs * self.world_size if self.distributed else bs
            chunk = indices[pos:pos + bs_global]
            if len(chunk) < bs_global:
                break  # drop ragged tail (train mode)
            pos += bs_global
            batches.append((seq_len, chunk))
        if self.shuffle:
            order = torch.randperm(len(batches), generator=g).tolist()
            batches = [batches[i] for i in order]
        self._batches = batches
        self._num_batches = len(batches)

    def set_epoch(self, epoch: int):
        if epoch != self.epoch:
            self.epoch = epoch
            self._update_schedule()

    def __len__(self):
        return self._num_batches

    def _make_batch(self, seq_len: int, sample_indices: List[int]):
        transform = self.transforms.get(seq_len)
        imgs, targets = [], []
        for idx in sample_indices:
            img, target = self.base_dataset[idx]
            if transform is not None:
                img = transform(img)
            imgs.append(img)
            targets.append(target)
        if self.mixup_fn is not None:
            # variable-size mixup runs on the pre-patchify image tensors and
            # returns provenance-matched soft targets (naflex_mixup.py)
            imgs, targets = self.mixup_fn(imgs, torch.tensor(targets))
        samples = [(self.patchifier(im), tg) for im, tg in zip(imgs, targets)]
        return self.collators[seq_len](samples)

    def __iter__(self) -> Iterator:
        worker_info = get_worker_info()
        wid = worker_info.id if worker_info else 0
        nworkers = worker_info.num_workers if worker_info else 1

        for bi, (seq_len, chunk) in enumerate(self._batches):
            if bi % nworkers != wid:
                continue
            if self.distributed:
                per_rank = len(chunk) // self.world_size
                chunk = chunk[self.rank * per_rank:(self.rank + 1) * per_rank]
            yield self._make_batch(seq_len, chunk)
