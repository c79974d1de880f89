"""Geometry transforms for the input pipeline.

Behavioral parity: /root/reference/timm/data/transforms.py
(RandomResizedCropAndInterpolation :166, CenterCropOrPad :314,
RandomCropOrPad :379, ResizeKeepRatio :448, TrimBorder :567, tensor/numpy
converters, str<->interpolation maps).  All image math routes through the
in-repo torchvision-free functional layer (`image_ops`).
"""
import math
import numbers
import random
import warnings
from typing import List, Sequence, Tuple, Union

import numpy as np
import torch
from PIL import Image

from . import image_ops as F
from .image_ops import InterpolationMode

has_interpolation_mode = True

__all__ = [
    "ToNumpy", "ToTensor", "str_to_interp_mode", "str_to_pil_interp", "interp_mode_to_str",
    "RandomResizedCropAndInterpolation", "CenterCropOrPad", "center_crop_or_pad", "crop_or_pad",
    "RandomCropOrPad", "RandomPad", "ResizeKeepRatio", "TrimBorder", "MaybeToTensor", "MaybePILToTensor"
]


# ---------------------------------------------------------------------------
# converters
# ---------------------------------------------------------------------------

class ToNumpy:
    """PIL -> CHW uint8 ndarray (no scaling); used by the fast-collate path."""

    def __call__(self, pil_img):
        arr = np.array(pil_img, dtype=np.uint8)
        if arr.ndim < 3:
            arr = arr[:, :, None]
        return np.rollaxis(arr, 2)


class ToTensor:
    """PIL -> tensor of the requested dtype, values NOT rescaled to [0,1]."""

    def __init__(self, dtype=torch.float32):
        self.dtype = dtype

    def __call__(self, pil_img):
        return F.pil_to_tensor(pil_img).to(dtype=self.dtype)


class MaybeToTensor(torch.nn.Module):
    """to_tensor (with [0,1] scaling) unless the input is already a tensor."""

    def forward(self, pic) -> torch.Tensor:
        return pic if isinstance(pic, torch.Tensor) else F.to_tensor(pic)

    def __repr__(self) -> str:
        return f"{self.__class__.__name__}()"


class MaybePILToTensor(torch.nn.Module):
    """pil_to_tensor (no value scaling) unless the input is already a tensor."""

    def forward(self, pic):
        return pic if isinstance(pic, torch.Tensor) else F.pil_to_tensor(pic)

    def __repr__(self) -> str:
        return f"{self.__class__.__name__}()"


# ---------------------------------------------------------------------------
# interpolation-mode naming (one triple table -> all direction maps)
# ---------------------------------------------------------------------------

_PIL_MODES = Image.Resampling if hasattr(Image, 'Resampling') else Image
_INTERP_TRIPLES = [
    # (name, PIL resample, torch InterpolationMode)
    ('nearest', _PIL_MODES.NEAREST, InterpolationMode.NEAREST),
    ('bilinear', _PIL_MODES.BILINEAR, InterpolationMode.BILINEAR),
    ('bicubic', _PIL_MODES.BICUBIC, InterpolationMode.BICUBIC),
    ('box', _PIL_MODES.BOX, InterpolationMode.BOX),
    ('hamming', _PIL_MODES.HAMMING, InterpolationMode.HAMMING),
    ('lanczos', _PIL_MODES.LANCZOS, InterpolationMode.LANCZOS),
]
_pil_interpolation_to_str = {pil: name for name, pil, _t in _INTERP_TRIPLES}
_str_to_pil_interpolation = {name: pil for name, pil, _t in _INTERP_TRIPLES}
_torch_interpolation_to_str = {t: name for name, _pil, t in _INTERP_TRIPLES}
_str_to_torch_interpolation = {name: t for name, _pil, t in _INTERP_TRIPLES}


def str_to_pil_interp(mode_str):
    return _str_to_pil_interpolation[mode_str]


def str_to_interp_mode(mode_str):
    return _str_to_torch_interpolation[mode_str]


def interp_mode_to_str(mode):
    return _torch_interpolation_to_str[mode]


_RANDOM_INTERPOLATION = (str_to_interp_mode('bilinear'), str_to_interp_mode('bicubic'))


def _resolve_interp(interpolation):
    """'random' -> the (bilinear, bicubic) pair; else the named mode."""
    if interpolation == 'random':
        return _RANDOM_INTERPOLATION
    return str_to_interp_mode(interpolation)


def _pick_interp(interpolation):
    if isinstance(interpolation, (tuple, list)):
        return random.choice(interpolation)
    return interpolation


def _interp_repr(interpolation):
    if isinstance(interpolation, (tuple, list)):
        return ' '.join(interp_mode_to_str(m) for m in interpolation)
    return interp_mode_to_str(interpolation)


def _setup_size(size, error_msg="Please provide only two dimensions (h, w) for size."):
    if isinstance(size, numbers.Number):
        return int(size), int(size)
    if isinstance(size, Sequence) and len(size) == 1:
        return size[0], size[0]
    if len(size) != 2:
        raise ValueError(error_msg)
    return size


# ---------------------------------------------------------------------------
# crops
# ---------------------------------------------------------------------------

class RandomResizedCropAndInterpolation:
    """Inception-style random-area/aspect crop resized to a fixed size, with
    per-call random interpolation when requested."""

    def __init__(
            self,
            size,
            scale=(0.08, 1.0),
            ratio=(3. / 4., 4. / 3.),
            interpolation='bilinear',
    ):
        self.size = tuple(size) if isinstance(size, (list, tuple)) else (size, size)
        if scale[0] > scale[1] or ratio[0] > ratio[1]:
            warnings.warn("range should be of kind (min, max)")
        self.interpolation = _resolve_interp(interpolation)
        self.scale = scale
        self.ratio = ratio

    @staticmethod
    def get_params(img, scale, ratio):
        """Sample a (top, left, h, w) crop box; central-crop fallback after
        10 rejected draws."""
        img_w, img_h = F.get_image_size(img)
        area = img_w * img_h
        log_ratio = (math.log(ratio[0]), math.log(ratio[1]))

        for _ in range(10):
            target_area = random.uniform(*scale) * area
            aspect = math.exp(random.uniform(*log_ratio))
            w = int(round(math.sqrt(target_area * aspect)))
            h = int(round(math.sqrt(target_area / aspect)))
            if w <= img_w and h <= img_h:
                top = random.randint(0, img_h - h)
                left = random.randint(0, img_w - w)
                return top, left, h, w

        # fallback: largest central crop within the aspect limits
        in_ratio = img_w / img_h
        if in_ratio < min(ratio):
            w = img_w
            h = int(round(w / min(ratio)))
        elif in_ratio > max(ratio):
            h = img_h
            w = int(round(h * max(ratio)))
        else:
            w, h = img_w, img_h
        return (img_h - h) // 2, (img_w - w) // 2, h, w

    def __call__(self, img):
        top, left, h, w = self.get_params(img, self.scale, self.ratio)
        return F.resized_crop(img, top, left, h, w, self.size, _pick_interp(self.interpolation))

    def __repr__(self):
        return (f'{self.__class__.__name__}(size={self.size}'
                f', scale={tuple(round(s, 4) for s in self.scale)}'
                f', ratio={tuple(round(r, 4) for r in self.ratio)}'
                f', interpolation={_interp_repr(self.interpolation)})')


def center_crop_or_pad(
        img: torch.Tensor,
        output_size: Union[int, List[int]],
        fill: Union[int, Tuple[int, int, int]] = 0,
        padding_mode: str = 'constant',
) -> torch.Tensor:
    """Center crop, symmetrically padding first when the target is larger."""
    crop_h, crop_w = _setup_size(output_size)
    _, img_h, img_w = F.get_dimensions(img)

    if crop_w > img_w or crop_h > img_h:
        extra_w = max(crop_w - img_w, 0)
        extra_h = max(crop_h - img_h, 0)
        img = F.pad(
            img,
            [extra_w // 2, extra_h // 2, (extra_w + 1) // 2, (extra_h + 1) // 2],
            fill=fill, padding_mode=padding_mode)
        _, img_h, img_w = F.get_dimensions(img)
        if (crop_w, crop_h) == (img_w, img_h):
            return img

    top = int(round((img_h - crop_h) / 2.0))
    left = int(round((img_w - crop_w) / 2.0))
    return F.crop(img, top, left, crop_h, crop_w)


class CenterCropOrPad(torch.nn.Module):
    """Module wrapper over center_crop_or_pad."""

    def __init__(
            self,
            size: Union[int, List[int]],
            fill: Union[int, Tuple[int, int, int]] = 0,
            padding_mode: str = 'constant',
    ):
        super().__init__()
        self.size = _setup_size(size)
        self.fill = fill
        self.padding_mode = padding_mode

    def forward(self, img):
        return center_crop_or_pad(img, self.size, fill=self.fill, padding_mode=self.padding_mode)

    def __repr__(self) -> str:
        return f"{self.__class__.__name__}(size={self.size})"


def crop_or_pad(
        img: torch.Tensor,
        top: int,
        left: int,
        height: int,
        width: int,
        fill: Union[int, Tuple[int, int, int]] = 0,
        padding_mode: str = 'constant',
) -> torch.Tensor:
    """Crop a (possibly out-of-bounds) box, padding whatever falls outside."""
    _, img_h, img_w = F.get_dimensions(img)
    right, bottom = left + width, top + height
    if left < 0 or top < 0 or right > img_w or bottom > img_h:
        img = F.pad(
            img,
            [
                max(-left + min(0, right), 0),
                max(-top + min(0, bottom), 0),
                max(right - max(img_w, left), 0),
                max(bottom - max(img_h, top), 0),
            ],
            fill=fill, padding_mode=padding_mode)
    return F.crop(img, max(top, 0), max(left, 0), height, width)


class RandomCropOrPad(torch.nn.Module):
    """Random placement within the available crop margin (or pad margin when
    the image is smaller than the target)."""

    def __init__(
            self,
            size: Union[int, List[int]],
            fill: Union[int, Tuple[int, int, int]] = 0,
            padding_mode: str = 'constant',
    ):
        super().__init__()
        self.size = _setup_size(size)
        self.fill = fill
        self.padding_mode = padding_mode

    @staticmethod
    def get_params(img, size):
        _, img_h, img_w = F.get_dimensions(img)
        dh, dw = img_h - size[0], img_w - size[1]
        top = int(math.copysign(random.randint(0, abs(dh)), dh))
        left = int(math.copysign(random.randint(0, abs(dw)), dw))
        return top, left

    def forward(self, img):
        top, left = self.get_params(img, self.size)
        return crop_or_pad(
            img, top=top, left=left, height=self.size[0], width=self.size[1],
            fill=self.fill, padding_mode=self.padding_mode)

    def __repr__(self) -> str:
        return f"{self.__class__.__name__}(size={self.size})"


class RandomPad:
    """Pad up to input_size with a random left/top split of the slack."""

    def __init__(self, input_size, fill=0):
        self.input_size = input_size
        self.fill = fill

    @staticmethod
    def get_params(img, input_size):
        w, h = F.get_image_size(img)
        slack_w = max(input_size[1] - w, 0)
        slack_h = max(input_size[0] - h, 0)
        pad_left = random.randint(0, slack_w)
        pad_top = random.randint(0, slack_h)
        return pad_left, pad_top, slack_w - pad_left, slack_h - pad_top

    def __call__(self, img):
        return F.pad(img, self.get_params(img, self.input_size), self.fill)


class ResizeKeepRatio:
    """Aspect-preserving resize toward a target box.

    ``longest`` interpolates between fit-shortest (0, may overshoot target)
    and fit-longest (1, always inside target); optional random scale/aspect
    jitter for train-time use."""

    def __init__(
            self,
            size,
            longest=0.,
            interpolation='bilinear',
            random_scale_prob=0.,
            random_scale_range=(0.85, 1.05),
            random_scale_area=False,
            random_aspect_prob=0.,
            random_aspect_range=(0.9, 1.11),
    ):
        self.size = tuple(size) if isinstance(size, (list, tuple)) else (size, size)
        self.interpolation = _resolve_interp(interpolation)
        self.longest = float(longest)
        self.random_scale_prob = random_scale_prob
        self.random_scale_range = random_scale_range
        self.random_scale_area = random_scale_area
        self.random_aspect_prob = random_aspect_prob
        self.random_aspect_range = random_aspect_range

    @staticmethod
    def get_params(
            img,
            target_size,
            longest,
            random_scale_prob=0.,
            random_scale_range=(1.0, 1.33),
            random_scale_area=False,
            random_aspect_prob=0.,
            random_aspect_range=(0.9, 1.11),
    ):
        """Compute the output (h, w)."""
        img_w, img_h = F.get_image_size(img)
        target_h, target_w = target_size
        # blend of shortest-fit and longest-fit downscale ratios
        rh, rw = img_h / target_h, img_w / target_w
        ratio = max(rh, rw) * longest + min(rh, rw) * (1. - longest)

        scale_w = scale_h = 1.
        if random_scale_prob > 0 and random.random() < random_scale_prob:
            jitter = random.uniform(random_scale_range[0], random_scale_range[1])
            if random_scale_area:
                # treat the draw as an area factor (RRC-like): <1 zooms in
                jitter = 1. / math.sqrt(jitter)
            scale_w = scale_h = jitter
        if random_aspect_prob > 0 and random.random() < random_aspect_prob:
            log_rng = (math.log(random_aspect_range[0]), math.log(random_aspect_range[1]))
            aspect = math.sqrt(math.exp(random.uniform(*log_rng)))
            # split the aspect jitter evenly across both dims
            scale_w, scale_h = scale_w / aspect, scale_h * aspect

        return [
            round(img_h * scale_h / ratio),
            round(img_w * scale_w / ratio),
        ]

    def __call__(self, img):
        size = self.get_params(
            img, self.size, self.longest,
            self.random_scale_prob, self.random_scale_range, self.random_scale_area,
            self.random_aspect_prob, self.random_aspect_range)
        return F.resize(img, size, _pick_interp(self.interpolation))

    def __repr__(self):
        return (f'{self.__class__.__name__}(size={self.size}'
                f', interpolation={_interp_repr(self.interpolation)}'
                f', longest={self.longest:.3f})')


class TrimBorder(torch.nn.Module):
    """Crop a fixed border from every edge."""

    def __init__(self, border_size: int):
        super().__init__()
        self.border_size = border_size

    def forward(self, img):
        w, h = F.get_image_size(img)
        trim = min(self.border_size, h)
        return F.crop(
            img, trim, trim,
            max(0, h - 2 * self.border_size),
            max(0, w - 2 * self.border_size))
