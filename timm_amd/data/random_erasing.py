"""Random Erasing (arxiv 1708.04896) — runs on-device inside the prefetcher,
after normalization, on the side HIP stream.

Behavioral parity: /root/reference/timm/data/random_erasing.py:26
(const/rand/pixel fill modes, count splitting, AugMix split skipping).
"""
import math
import random

import torch

__all__ = ['RandomErasing']


class RandomErasing:
    """Erase random rectangles from (normalized) image tensors.

    Fill modes: 'const' zeros · 'rand' one normal draw per block channel ·
    'pixel' per-pixel normal noise.  With num_splits > 1 the first (clean)
    split of an AugMix batch is left untouched.
    """

    def __init__(
            self,
            probability=0.5,
            min_area=0.02,
            max_area=1 / 3,
            min_aspect=0.3,
            max_aspect=None,
            mode='const',
            min_count=1,
            max_count=None,
            num_splits=0,
            device='cuda',
    ):
        self.probability = probability
        self.min_area = min_area
        self.max_area = max_area
        max_aspect = max_aspect or 1 / min_aspect
        self.log_aspect_ratio = (math.log(min_aspect), math.log(max_aspect))
        self.min_count = min_count
        self.max_count = max_count or min_count
        self.num_splits = num_splits
        self.mode = mode.lower()
        assert self.mode in ('', 'const', 'rand', 'pixel')
        self.device = device

    def _fill(self, shape, dtype):
        chan, h, w = shape
        if self.mode == 'pixel':
            return torch.empty(shape, dtype=dtype, device=self.device).normal_()
        if self.mode == 'rand':
            return torch.empty((chan, 1, 1), dtype=dtype, device=self.device).normal_()
        return torch.zeros((chan, 1, 1), dtype=dtype, device=self.device)

    def _sample_box(self, img_h, img_w, count):
        """One rejection-sampled (top, left, h, w) box, or None."""
        for _ in range(10):
            area = random.uniform(self.min_area, self.max_area) * img_h * img_w / count
            aspect = math.exp(random.uniform(*self.log_aspect_ratio))
            h = int(round(math.sqrt(area * aspect)))
            w = int(round(math.sqrt(area / aspect)))
            if w < img_w and h < img_h:
                return random.randint(0, img_h - h), random.randint(0, img_w - w), h, w
        return None

    def _erase(self, img, chan, img_h, img_w, dtype):
        if random.random() > self.probability:
            return
        count = self.min_count if self.min_count == self.max_count else \
            random.randint(self.min_count, self.max_count)
        for _ in range(count):
            box = self._sample_box(img_h, img_w, count)
            if box is not None:
                top, left, h, w = box
                img[:, top:top + h, left:left + w] = self._fill((chan, h, w), dtype)

    def __call__(self, x):
        if x.ndim == 3:
            self._erase(x, *x.shape, x.dtype)
            return x
        batch_size = x.shape[0]
        start = batch_size // self.num_splits if self.num_splits > 1 else 0
        for i in range(start, batch_size):
            self._erase(x[i], *x.shape[1:], x.dtype)
        return x

    def __repr__(self):
        return (f'{self.__class__.__name__}(p={self.probability}, mode={self.mode}'
                f', count=({self.min_count}, {self.max_count}))')
