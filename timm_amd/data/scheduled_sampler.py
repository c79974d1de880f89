"""Scheduled batch sampling + per-batch transform dispatch (reference
`timm/data/scheduled_sampler.py:11-331`).

Used for scheduled-resolution training: every batch carries a "choice index"
that selects both its batch size and its transform (image resolution). The
schedule is derived only from (seed, epoch, sampler length), so every
distributed rank builds the identical batch-shape sequence without any
communication — batches stay shape-aligned across ranks for the bucketed
all-reduce.

Two modes:
* sample-budget (num_batches None): compose (choice, batch_size) pairs until
  the sampler is exhausted, fixed composition re-shuffled each epoch.
* fixed-batch (num_batches set, or progressive schedule): draw num_batches
  choices per epoch from the (possibly progressive) choice distribution and
  cycle the index stream as needed.
"""
import math
from itertools import islice
from typing import Any, Callable, Iterator, List, Optional, Sequence, Tuple, Union

import torch
from torch.utils.data import Dataset, Sampler


class ScheduledBatchSampler(Sampler):
    """Yields lists of (sample_index, choice_index) pairs; pair batches with
    `ScheduledTransformDataset` so the choice picks the transform."""

    def __init__(
            self,
            sampler: Sampler,
            batch_sizes: Sequence[int],
            choice_weights: Optional[Sequence[float]] = None,
            seed: int = 0,
            drop_last: bool = True,
            shuffle_schedule: bool = True,
            num_batches: Optional[int] = None,
            choice_schedule: str = 'constant',
            schedule_epochs: Optional[int] = None,
            schedule_spread: float = 0.65,
            schedule_random_mix: float = 0.1,
    ) -> None:
        if not hasattr(sampler, '__len__'):
            raise TypeError('ScheduledBatchSampler requires a sampler with a length.')
        if len(sampler) <= 0:
            raise ValueError('ScheduledBatchSampler requires a non-empty sampler.')
        if not batch_sizes:
            raise ValueError('batch_sizes must contain at least one value.')
        if any(int(b) != b or b <= 0 for b in batch_sizes):
            raise ValueError('All scheduled batch sizes must be positive integers.')
        if num_batches is not None and (int(num_batches) != num_batches or num_batches <= 0):
            raise ValueError('num_batches must be a positive integer when specified.')
        if choice_schedule not in ('constant', 'progressive'):
            raise ValueError("choice_schedule must be 'constant' or 'progressive'.")
        if choice_schedule == 'progressive':
            if len(batch_sizes) < 2:
                raise ValueError('A progressive schedule requires at least two choices.')
            if schedule_epochs is None or int(schedule_epochs) != schedule_epochs or schedule_epochs <= 0:
                raise ValueError('schedule_epochs must be a positive integer for a progressive schedule.')
            if schedule_spread < 0:
                raise ValueError('schedule_spread must be non-negative.')
            if not 0 <= schedule_random_mix <= 1:
                raise ValueError('schedule_random_mix must be between 0 and 1.')

        self.sampler = sampler
        self.batch_sizes = tuple(int(b) for b in batch_sizes)
        self.choice_weights = self._normalize_choice_weights(choice_weights)
        self._active_choices = tuple(
            i for i, w in enumerate(self.choice_weights) if w > 0)
        self.seed = seed
        self.drop_last = drop_last
        self.shuffle_schedule = shuffle_schedule
        self.choice_schedule = choice_schedule
        self.schedule_epochs = int(schedule_epochs) if schedule_epochs is not None else None
        self.schedule_spread = schedule_spread
        self.schedule_random_mix = schedule_random_mix
        self.epoch = 0
        self.average_batch_size = self._calculate_average_batch_size()
        if choice_schedule == 'progressive' and num_batches is None:
            num_batches = self._infer_num_batches()
        self.num_batches = int(num_batches) if num_batches is not None else None
        self._sample_budget_schedule: Tuple[Tuple[int, int], ...] = ()
        if self.num_batches is None:
            self._sample_budget_schedule = self._create_sample_budget_schedule()
            if not self._sample_budget_schedule:
                raise ValueError('No full scheduled batch fits the sampler; reduce the batch sizes.')

    def _normalize_choice_weights(self, choice_weights: Optional[Sequence[float]]) -> torch.Tensor:
        n = len(self.batch_sizes)
        if choice_weights is None:
            return torch.full((n,), 1.0 / n, dtype=torch.float64)
        if len(choice_weights) != n:
            raise ValueError('choice_weights and batch_sizes must have the same length.')
        weights = torch.tensor(choice_weights, dtype=torch.float64)
        if not torch.isfinite(weights).all() or (weights < 0).any():
            raise ValueError('choice_weights must contain finite, non-negative values.')
        total = weights.sum()
        if total <= 0:
            raise ValueError('choice_weights must have a positive sum.')
        return weights / total

    def choice_weights_for_epoch(self, epoch: int) -> torch.Tensor:
        """Normalized choice weights for an epoch; in progressive mode a
        Gaussian probability window slides from the first choice to the last."""
        if self.choice_schedule == 'constant' or len(self.batch_sizes) == 1:
            return self.choice_weights

        if self.schedule_epochs == 1:
            progress = 1.0
        else:
            progress = min(max(epoch / (self.schedule_epochs - 1), 0.0), 1.0)
        positions = torch.arange(len(self.batch_sizes), dtype=torch.float64)
        center = progress * (len(self.batch_sizes) - 1)

        if self.schedule_spread == 0:
            distances = (positions - center).abs()
            weights = (distances == distances.min()).to(torch.float64)
        else:
            weights = torch.exp(-0.5 * ((positions - center) / self.schedule_spread) ** 2)
        weights *= self.choice_weights
        if weights.sum() <= 0:
            nearest = min(self._active_choices, key=lambda i: abs(i - center))
            weights = torch.zeros_like(self.choice_weights)
            weights[nearest] = 1.0
        else:
            weights /= weights.sum()

        if self.schedule_random_mix:
            uniform = (self.choice_weights > 0).to(weights.dtype)
            uniform /= uniform.sum()
            weights = (1.0 - self.schedule_random_mix) * weights + self.schedule_random_mix * uniform
        return weights / weights.sum()

    def _calculate_average_batch_size(self) -> float:
        sizes = torch.tensor(self.batch_sizes, dtype=torch.float64)
        if self.choice_schedule == 'progressive':
            per_epoch = [
                torch.dot(self.choice_weights_for_epoch(e), sizes) for e in range(self.schedule_epochs)]
            return float(torch.stack(per_epoch).mean().item())
        return float(torch.dot(self.choice_weights, sizes).item())

    def _infer_num_batches(self) -> int:
        if len(set(self.batch_sizes)) == 1:
            b = self.batch_sizes[0]
            n = len(self.sampler) // b if self.drop_last else math.ceil(len(self.sampler) / b)
        else:
            if self.drop_last:
                n = int(len(self.sampler) / self.average_batch_size)
            else:
                n = math.ceil(len(self.sampler) / self.average_batch_size)
        if n < 1:
            raise ValueError('No full scheduled batch fits the sampler; reduce the batch sizes.')
        return n

    def _sample_choice(
            self,
            generator: torch.Generator,
            valid_choices: Optional[Sequence[int]] = None,
            choice_weights: Optional[torch.Tensor] = None,
    ) -> int:
        choice_weights = self.choice_weights if choice_weights is None else choice_weights
        if valid_choices is None:
            return int(torch.multinomial(choice_weights, 1, generator=generator).item())
        valid_choices = tuple(valid_choices)
        weights = choice_weights[list(valid_choices)]
        if weights.sum() <= 0:
            raise RuntimeError('No positive-weight scheduled choice is available for this batch.')
        picked = int(torch.multinomial(weights, 1, generator=generator).item())
        return valid_choices[picked]

    def _create_sample_budget_schedule(self) -> Tuple[Tuple[int, int], ...]:
        generator = torch.Generator().manual_seed(self.seed)
        remaining = len(self.sampler)
        min_batch = min(self.batch_sizes[i] for i in self._active_choices)
        schedule = []
        while remaining >= min_batch:
            valid = [i for i in self._active_choices if self.batch_sizes[i] <= remaining]
            choice = self._sample_choice(generator, valid)
            schedule.append((choice, self.batch_sizes[choice]))
            remaining -= self.batch_sizes[choice]
        if remaining and not self.drop_last:
            schedule.append((self._sample_choice(generator), remaining))
        return tuple(schedule)

    def _create_fixed_batch_schedule(self, epoch: int) -> Tuple[Tuple[int, int], ...]:
        generator = torch.Generator().manual_seed(self.seed + 2 * epoch)
        weights = self.choice_weights_for_epoch(epoch)
        schedule = []
        for _ in range(self.num_batches):
            choice = self._sample_choice(generator, choice_weights=weights)
            schedule.append((choice, self.batch_sizes[choice]))
        return tuple(schedule)

    def _create_schedule(self, epoch: int) -> Tuple[Tuple[int, int], ...]:
        if self.num_batches is not None:
            return self._create_fixed_batch_schedule(epoch)
        return self._sample_budget_schedule

    @property
    def schedule(self) -> Tuple[Tuple[int, int], ...]:
        """Unshuffled schedule for the currently selected epoch."""
        return self._create_schedule(self.epoch)

    def _cycling_sampler(self) -> Iterator[Any]:
        while True:
            yielded = False
            for sample_index in self.sampler:
                yielded = True
                yield sample_index
            if not yielded:
                break

    def __iter__(self) -> Iterator[List[Tuple[Any, int]]]:
        epoch = self.epoch
        schedule = self._create_schedule(epoch)
        if self.shuffle_schedule and len(schedule) > 1:
            generator = torch.Generator().manual_seed(self.seed + 2 * epoch + 1)
            order = torch.randperm(len(schedule), generator=generator).tolist()
            schedule = tuple(schedule[i] for i in order)

        indices = self._cycling_sampler() if self.num_batches is not None else iter(self.sampler)
        for choice, batch_size in schedule:
            batch = list(islice(indices, batch_size))
            if len(batch) != batch_size:
                if self.drop_last or not batch:
                    break
            yield [(sample_index, choice) for sample_index in batch]

    def __len__(self) -> int:
        if self.num_batches is not None:
            return self.num_batches
        return len(self._sample_budget_schedule)

    def set_epoch(self, epoch: int) -> None:
        self.epoch = epoch
        if hasattr(self.sampler, 'set_epoch'):
            self.sampler.set_epoch(epoch)


class ScheduledTransformDataset(Dataset):
    """Map-style wrapper: index arrives as (sample_index, choice_index) from
    `ScheduledBatchSampler`; the choice selects which transform to apply."""

    def __init__(self, dataset: Dataset, transforms: Sequence[Callable]) -> None:
        if not transforms:
            raise ValueError('transforms must contain at least one transform.')
        self.dataset = dataset
        self.transforms = tuple(transforms)

    def __getitem__(self, scheduled_index: Tuple[Any, int]) -> Union[Tuple[Any, ...], List[Any]]:
        sample_index, transform_index = scheduled_index
        if not 0 <= transform_index < len(self.transforms):
            raise IndexError(f'Transform index {transform_index} is out of range.')
        sample = self.dataset[sample_index]
        if not isinstance(sample, (tuple, list)) or not sample:
            raise TypeError('ScheduledTransformDataset expects tuple/list dataset samples.')
        image = self.transforms[transform_index](sample[0])
        if isinstance(sample, tuple):
            return (image, *sample[1:])
        return [image, *sample[1:]]

    def __len__(self) -> int:
        return len(self.dataset)

    def set_epoch(self, epoch: int) -> None:
        if hasattr(self.dataset, 'set_epoch'):
            self.dataset.set_epoch(epoch)

    def filename(self, index: int, basename: bool = False, absolute: bool = False) -> Any:
        return self.dataset.filename(index, basename=basename, absolute=absolute)

    def filenames(self, basename: bool = False, absolute: bool = False) -> Any:
        return self.dataset.filenames(basename=basename, absolute=absolute)
