"""Distributed samplers: ordered eval sharding + repeated-augmentation.

Behavioral parity: /root/reference/timm/data/distributed_sampler.py:7,54
(same pad-to-equal + strided rank subsample index streams).
"""
import math

import torch
import torch.distributed as dist
from torch.utils.data import Sampler

__all__ = ['OrderedDistributedSampler', 'RepeatAugSampler']


def _rank_world(num_replicas, rank):
    if num_replicas is None or rank is None:
        if not dist.is_available():
            raise RuntimeError('Requires distributed package to be available')
        num_replicas = dist.get_world_size() if num_replicas is None else num_replicas
        rank = dist.get_rank() if rank is None else rank
    return num_replicas, rank


def _pad_and_shard(indices, total_size, rank, num_replicas):
    """Wrap-pad ``indices`` to ``total_size`` then take this rank's stride."""
    short = total_size - len(indices)
    if short > 0:
        indices = indices + indices[:short]
    assert len(indices) == total_size
    shard = indices[rank:total_size:num_replicas]
    return shard


class OrderedDistributedSampler(Sampler):
    """In-order rank sharding with wrap padding to equal per-rank length.

    Used for distributed eval where sample order must be stable (no shuffle)
    and every rank must run the same number of batches.
    """

    def __init__(self, dataset, num_replicas=None, rank=None):
        self.dataset = dataset
        self.num_replicas, self.rank = _rank_world(num_replicas, rank)
        self.num_samples = math.ceil(len(dataset) / self.num_replicas)
        self.total_size = self.num_samples * self.num_replicas

    def __iter__(self):
        shard = _pad_and_shard(
            list(range(len(self.dataset))), self.total_size, self.rank, self.num_replicas)
        assert len(shard) == self.num_samples
        return iter(shard)

    def __len__(self):
        return self.num_samples


class RepeatAugSampler(Sampler):
    """Repeated-augmentation sampling (arxiv 1902.05509 / DeiT RASampler).

    Each sample index is repeated ``num_repeats`` times; the strided rank
    shard then routes different repeats of the same image to different ranks
    so each GPU sees an independently-augmented copy.  Per-epoch shuffling is
    seeded by ``set_epoch``.
    """

    def __init__(
            self,
            dataset,
            num_replicas=None,
            rank=None,
            shuffle=True,
            num_repeats=3,
            selected_round=256,
            selected_ratio=0,
    ):
        self.dataset = dataset
        self.num_replicas, self.rank = _rank_world(num_replicas, rank)
        self.shuffle = shuffle
        self.num_repeats = num_repeats
        self.epoch = 0
        self.num_samples = math.ceil(len(dataset) * num_repeats / self.num_replicas)
        self.total_size = self.num_samples * self.num_replicas
        # per-epoch this rank yields ~len(dataset)/ratio samples, optionally
        # floored to a multiple of selected_round (matches DeiT defaults)
        ratio = selected_ratio or self.num_replicas
        if selected_round:
            self.num_selected_samples = int(
                len(dataset) // selected_round * selected_round / ratio)
        else:
            self.num_selected_samples = math.ceil(len(dataset) / ratio)

    def set_epoch(self, epoch):
        self.epoch = epoch

    def _repeat(self, indices: torch.Tensor) -> list:
        reps = self.num_repeats
        if isinstance(reps, float) and not reps.is_integer():
            # fractional repeat factor: index-map resample
            out_len = math.ceil(reps * len(self.dataset))
            picks = torch.tensor([int(i // reps) for i in range(out_len)])
            return indices[picks].tolist()
        return torch.repeat_interleave(indices, repeats=int(reps), dim=0).tolist()

    def __iter__(self):
        g = torch.Generator()
        g.manual_seed(self.epoch)
        if self.shuffle:
            order = torch.randperm(len(self.dataset), generator=g)
        else:
            order = torch.arange(len(self.dataset))
        shard = _pad_and_shard(self._repeat(order), self.total_size, self.rank, self.num_replicas)
        assert len(shard) == self.num_samples
        return iter(shard[:self.num_selected_samples])

    def __len__(self):
        return self.num_selected_samples
