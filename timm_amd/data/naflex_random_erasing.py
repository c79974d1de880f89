"""Patch-granular random erasing for NaFlex (patchified) inputs.

Behavioral parity: /root/reference/timm/data/naflex_random_erasing.py.
Runs on-device inside the NaFlex prefetcher; two strategies:
  'patch'  — erase independently-sampled valid patches (speckle noise)
  'region' — erase a contiguous patch-aligned rectangle (classic RE), using
             the patch coordinate grid rather than pixel coordinates.
"""
import math
import random
from typing import Optional, Tuple, Union

import torch

__all__ = ['PatchRandomErasing']


class PatchRandomErasing:
    """Random erasing over [B, N, P*P, C] (or [B, N, Ph, Pw, C]) patch tensors."""

    def __init__(
            self,
            erase_prob: float = 0.5,
            patch_drop_prob: float = 0.0,
            min_count: int = 1,
            max_count: Optional[int] = None,
            min_area: float = 0.02,
            max_area: float = 1 / 3,
            min_aspect: float = 0.3,
            max_aspect: Optional[float] = None,
            mode: str = 'const',
            value: float = 0.,
            spatial_mode: str = 'region',
            num_splits: int = 0,
            device: Union[str, torch.device] = 'cuda',
    ) -> None:
        self.erase_prob = erase_prob
        self.patch_drop_prob = patch_drop_prob
        self.min_count = min_count
        self.max_count = max_count or min_count
        self.min_area = min_area
        self.max_area = max_area
        max_aspect = max_aspect or 1 / min_aspect
        self.log_aspect_ratio = (math.log(min_aspect), math.log(max_aspect))
        self.num_splits = num_splits
        self.device = device
        assert spatial_mode in ('patch', 'region')
        self.spatial_mode = spatial_mode
        self.erase_mode = mode.lower()
        assert self.erase_mode in ('rand', 'pixel', 'const')
        self.const_value = value
        self.unique_noise_per_patch = True

    # -- fill values --------------------------------------------------------
    def _fill(self, shape, dtype, device, value=None):
        """Erase content: per-pixel noise ('pixel'), broadcast noise ('rand'),
        or a constant."""
        if self.erase_mode == 'pixel':
            return torch.empty(shape, dtype=dtype, device=device).normal_()
        bshape = (1, 1, shape[-1]) if len(shape) == 3 else (1, shape[-1])
        if self.erase_mode == 'const' or value is not None:
            fill = self.const_value if value is None else value
            if isinstance(fill, (int, float)):
                return torch.full(bshape, fill, dtype=dtype, device=device)
            return torch.tensor(fill, dtype=dtype, device=device).expand(bshape).clone()
        return torch.empty(bshape, dtype=dtype, device=device).normal_()

    def _noise_shape(self, n_selected, patch_shape):
        if self.unique_noise_per_patch and self.erase_mode == 'pixel':
            return (n_selected,) + tuple(patch_shape)
        return tuple(patch_shape)

    # -- strategies (operate on one sample, in place) ------------------------
    def _erase_patches(self, patches, patch_coord, patch_valid, patch_shape, dtype):
        """Erase a random subset of valid patches."""
        if random.random() > self.erase_prob:
            return
        valid_idx = torch.nonzero(patch_valid, as_tuple=True)[0]
        n_valid = len(valid_idx)
        if n_valid == 0:
            return
        count = random.randint(self.min_count, self.max_count)
        hi = min(n_valid, max(1, int(n_valid * count * self.max_area)))
        lo = max(1, int(n_valid * count * self.min_area))
        n_erase = random.randint(lo, hi)
        chosen = valid_idx[torch.randperm(n_valid, device=patches.device)[:n_erase]]
        patches[chosen] = self._fill(
            self._noise_shape(n_erase, patch_shape), dtype, patches.device)

    def _erase_region(self, patches, patch_coord, patch_valid, patch_shape, dtype):
        """Erase patch-aligned rectangles on the coordinate grid."""
        if random.random() > self.erase_prob:
            return
        valid_coord = patch_coord[patch_valid]
        if len(valid_coord) == 0:
            return
        grid_h = valid_coord[:, 0].max().item() + 1
        grid_w = valid_coord[:, 1].max().item() + 1
        ys, xs = patch_coord[:, 0], patch_coord[:, 1]

        for _ in range(random.randint(self.min_count, self.max_count)):
            for _attempt in range(10):
                area = random.uniform(self.min_area, self.max_area) * grid_h * grid_w
                aspect = math.exp(random.uniform(*self.log_aspect_ratio))
                h = int(round(math.sqrt(area * aspect)))
                w = int(round(math.sqrt(area / aspect)))
                if h > grid_h or w > grid_w:
                    continue
                top = random.randint(0, grid_h - h)
                left = random.randint(0, grid_w - w)
                inside = (
                    (ys >= top) & (ys < top + h) &
                    (xs >= left) & (xs < left + w) &
                    patch_valid
                )
                n_inside = int(inside.sum().item())
                if not n_inside:
                    continue
                patches[inside] = self._fill(
                    self._noise_shape(n_inside, patch_shape), dtype, patches.device)
                break

    # -- entry ---------------------------------------------------------------
    def __call__(
            self,
            patches: torch.Tensor,
            patch_coord: torch.Tensor,
            patch_valid: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        assert patches.ndim in (4, 5), 'expected [B,N,P*P,C] or [B,N,Ph,Pw,C]'
        B, N = patches.shape[:2]
        patch_shape = patches.shape[2:]
        if patch_valid is None:
            patch_valid = torch.ones((B, N), dtype=torch.bool, device=patches.device)

        # AugMix-style split handling: leave the clean split untouched
        start = B // self.num_splits if self.num_splits > 1 else 0
        erase = self._erase_patches if self.spatial_mode == 'patch' else self._erase_region
        for i in range(start, B):
            if self.patch_drop_prob:
                raise NotImplementedError(
                    'patch dropout requires non-contiguous patch support in the model')
            erase(patches[i], patch_coord[i], patch_valid[i], patch_shape, patches.dtype)
        return patches

    def __repr__(self) -> str:
        return (f'{self.__class__.__name__}(p={self.erase_prob}, mode={self.erase_mode}'
                f', spatial={self.spatial_mode}, area=({self.min_area}, {self.max_area}))'
                f', count=({self.min_count}, {self.max_count}))')
