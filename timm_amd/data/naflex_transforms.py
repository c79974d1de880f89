"""NaFlex variable-resolution transforms (reference `timm/data/naflex_transforms.py`).

`get_image_size_for_seq` (:26), `RandomResizedCropToSequence` (:496),
`patchify_image` (:751), `Patchify` (:807).
"""
import math
import random
from typing import List, Optional, Tuple, Union

import numpy as np
import torch
from PIL import Image

from . import image_ops as F
from .image_ops import InterpolationMode


def get_image_size_for_seq(
        image_hw: Tuple[int, int],
        patch_size: int = 16,
        max_seq_len: int = 1024,
        divisible_by_patch: bool = True,
        max_ratio: Optional[float] = None,
) -> Tuple[float, Tuple[int, int]]:
    """Determine scaling ratio + target size so the patchified seq fits max_seq_len
    while preserving aspect (reference `:26`)."""
    h, w = image_hw

    def seq_len_for(scale):
        th = max(1, math.ceil(h * scale / patch_size)) if divisible_by_patch else max(1, (int(h * scale) + patch_size - 1) // patch_size)
        tw = max(1, math.ceil(w * scale / patch_size)) if divisible_by_patch else max(1, (int(w * scale) + patch_size - 1) // patch_size)
        return th * tw

    # binary search the largest scale with seq_len <= max_seq_len
    lo, hi = 1e-3, 10.0
    for _ in range(40):
        mid = (lo + hi) / 2
        if seq_len_for(mid) <= max_seq_len:
            lo = mid
        else:
            hi = mid
    ratio = lo
    if max_ratio is not None:
        ratio = min(ratio, max_ratio)

    nh = max(1, round(h * ratio / patch_size)) * patch_size
    nw = max(1, round(w * ratio / patch_size)) * patch_size
    # clamp once more in case rounding pushed over
    while (nh // patch_size) * (nw // patch_size) > max_seq_len:
        if nh >= nw:
            nh -= patch_size
        else:
            nw -= patch_size
    return ratio, (nh, nw)


class ResizeToSequence:
    """Resize image (preserving aspect) so its patch grid fits a sequence length."""

    def __init__(
            self,
            patch_size: int,
            max_seq_len: int = 1024,
            divisible_by_patch: bool = True,
            max_ratio: Optional[float] = None,
            interpolation: str = 'bicubic',
    ):
        self.patch_size = patch_size
        self.max_seq_len = max_seq_len
        self.divisible_by_patch = divisible_by_patch
        self.max_ratio = max_ratio
        self.interpolation = interpolation

    def __call__(self, img):
        w, h = F.get_image_size(img)
        _, (nh, nw) = get_image_size_for_seq(
            (h, w), self.patch_size, self.max_seq_len,
            divisible_by_patch=self.divisible_by_patch, max_ratio=self.max_ratio)
        return F.resize(img, (nh, nw), InterpolationMode(self.interpolation))


class RandomResizedCropToSequence:
    """Random-resized-crop whose output size targets a token budget while
    randomizing scale/aspect (reference `:496`)."""

    def __init__(
            self,
            patch_size: int,
            max_seq_len: int = 1024,
            scale: Tuple[float, float] = (0.08, 1.0),
            ratio: Tuple[float, float] = (3. / 4., 4. / 3.),
            divisible_by_patch: bool = True,
            max_ratio: Optional[float] = None,
            interpolation: str = 'bicubic',
    ):
        self.patch_size = patch_size
        self.max_seq_len = max_seq_len
        self.scale = scale
        self.ratio = ratio
        self.divisible_by_patch = divisible_by_patch
        self.max_ratio = max_ratio
        self.interpolation = interpolation

    def __call__(self, img):
        w, h = F.get_image_size(img)
        area = h * w
        for _ in range(10):
            target_area = random.uniform(*self.scale) * area
            log_ratio = (math.log(self.ratio[0]), math.log(self.ratio[1]))
            aspect = math.exp(random.uniform(*log_ratio))
            cw = int(round(math.sqrt(target_area * aspect)))
            ch = int(round(math.sqrt(target_area / aspect)))
            if cw <= w and ch <= h:
                top = random.randint(0, h - ch)
                left = random.randint(0, w - cw)
                img_c = F.crop(img, top, left, ch, cw)
                break
        else:
            img_c = img
            ch, cw = h, w
        _, (nh, nw) = get_image_size_for_seq(
            (ch, cw), self.patch_size, self.max_seq_len,
            divisible_by_patch=self.divisible_by_patch, max_ratio=self.max_ratio)
        return F.resize(img_c, (nh, nw), InterpolationMode(self.interpolation))


def patchify_image(
        img: torch.Tensor,
        patch_size: Tuple[int, int],
        pad: bool = True,
        flatten_patches: bool = True,
) -> Tuple[torch.Tensor, Tuple[int, int]]:
    """[C,H,W] tensor -> ([N, ph*pw*C] patches (P-P-C layout), (nh, nw))
    (reference `:751`)."""
    c, h, w = img.shape
    ph, pw = patch_size
    if pad and (h % ph or w % pw):
        import torch.nn.functional as TF
        img = TF.pad(img, (0, (pw - w % pw) % pw, 0, (ph - h % ph) % ph))
        c, h, w = img.shape
    nh, nw = h // ph, w // pw
    patches = img.view(c, nh, ph, nw, pw).permute(1, 3, 2, 4, 0)  # [nh, nw, ph, pw, c]
    if flatten_patches:
        patches = patches.reshape(nh * nw, ph * pw * c)
    else:
        patches = patches.reshape(nh * nw, ph, pw, c)
    return patches, (nh, nw)


class Patchify:
    """Transform: image tensor -> dict {patches, patch_coord, patch_valid}
    (reference `:807`)."""

    def __init__(self, patch_size: Union[int, Tuple[int, int]], flatten_patches: bool = True):
        self.patch_size = (patch_size, patch_size) if isinstance(patch_size, int) else tuple(patch_size)
        self.flatten_patches = flatten_patches

    def __call__(self, img):
        if isinstance(img, Image.Image):
            img = F.to_tensor(img)
        patches, (nh, nw) = patchify_image(img, self.patch_size, flatten_patches=self.flatten_patches)
        coord = torch.stack(torch.meshgrid(
            torch.arange(nh), torch.arange(nw), indexing='ij'), dim=-1).reshape(nh * nw, 2)
        return {
            'patches': patches,
            'patch_coord': coord,
            'patch_valid': torch.ones(nh * nw, dtype=torch.bool),
        }
