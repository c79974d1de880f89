"""Self-contained image transform primitives (no torchvision dependency).

Functional ops + transform classes covering the subset of
torchvision.transforms the pipeline needs: Compose, Resize, CenterCrop,
flips, ColorJitter, RandomApply, RandomGrayscale, GaussianBlur, Normalize,
ToTensor, and the functional resize/crop/pad/resized_crop used by the
geometry transforms.  PIL on the host (loader workers); tensor paths where
cheap.
"""
import math
import numbers
import random
from enum import Enum
from typing import List, Optional, Sequence, Tuple, Union

import numpy as np
import torch
from PIL import Image, ImageEnhance, ImageFilter, ImageOps


class InterpolationMode(str, Enum):
    NEAREST = 'nearest'
    BILINEAR = 'bilinear'
    BICUBIC = 'bicubic'
    BOX = 'box'
    HAMMING = 'hamming'
    LANCZOS = 'lanczos'


if hasattr(Image, 'Resampling'):
    _PIL_RESAMPLE = {
        InterpolationMode.NEAREST: Image.Resampling.NEAREST,
        InterpolationMode.BILINEAR: Image.Resampling.BILINEAR,
        InterpolationMode.BICUBIC: Image.Resampling.BICUBIC,
        InterpolationMode.BOX: Image.Resampling.BOX,
        InterpolationMode.HAMMING: Image.Resampling.HAMMING,
        InterpolationMode.LANCZOS: Image.Resampling.LANCZOS,
    }
else:
    _PIL_RESAMPLE = {
        InterpolationMode.NEAREST: Image.NEAREST,
        InterpolationMode.BILINEAR: Image.BILINEAR,
        InterpolationMode.BICUBIC: Image.BICUBIC,
        InterpolationMode.BOX: Image.BOX,
        InterpolationMode.HAMMING: Image.HAMMING,
        InterpolationMode.LANCZOS: Image.LANCZOS,
    }


def _to_interp(interpolation) -> InterpolationMode:
    if isinstance(interpolation, InterpolationMode):
        return interpolation
    if isinstance(interpolation, str):
        return InterpolationMode(interpolation)
    # PIL constant
    for k, v in _PIL_RESAMPLE.items():
        if v == interpolation:
            return k
    return InterpolationMode.BILINEAR


# ---------------------------------------------------------------------------
# functional API
# ---------------------------------------------------------------------------

def get_image_size(img) -> List[int]:
    """Returns [width, height]."""
    if isinstance(img, torch.Tensor):
        return [img.shape[-1], img.shape[-2]]
    return list(img.size)


def get_dimensions(img) -> List[int]:
    """Returns [channels, height, width]."""
    if isinstance(img, torch.Tensor):
        if img.ndim == 2:
            return [1, img.shape[0], img.shape[1]]
        return [img.shape[-3], img.shape[-2], img.shape[-1]]
    return [len(img.getbands()), img.size[1], img.size[0]]


def _setup_size2(size):
    if isinstance(size, numbers.Number):
        return int(size), int(size)
    if isinstance(size, Sequence) and len(size) == 1:
        return int(size[0]), int(size[0])
    return int(size[0]), int(size[1])


def resize(img, size, interpolation=InterpolationMode.BILINEAR, antialias: Optional[bool] = True):
    """Resize; scalar size == shortest-edge mode, (h, w) == exact."""
    interp = _to_interp(interpolation)
    if isinstance(img, torch.Tensor):
        import torch.nn.functional as TF
        if isinstance(size, int) or (isinstance(size, Sequence) and len(size) == 1):
            s = size if isinstance(size, int) else size[0]
            h, w = img.shape[-2:]
            if w <= h:
                ow, oh = s, max(1, int(s * h / w))
            else:
                oh, ow = s, max(1, int(s * w / h))
        else:
            oh, ow = int(size[0]), int(size[1])
        mode = {'nearest': 'nearest', 'bilinear': 'bilinear', 'bicubic': 'bicubic'}.get(interp.value, 'bilinear')
        x = img.unsqueeze(0) if img.ndim == 3 else img
        dtype = x.dtype
        if not x.is_floating_point():
            x = x.float()
        aa = antialias if mode in ('bilinear', 'bicubic') else False
        x = TF.interpolate(x, size=(oh, ow), mode=mode, antialias=aa,
                           align_corners=False if mode in ('bilinear', 'bicubic') else None)
        if not torch.is_floating_point(torch.empty(0, dtype=dtype)):
            x = x.round_().clamp_(0, 255).to(dtype)
        return x.squeeze(0) if img.ndim == 3 else x
    # PIL
    if isinstance(size, int) or (isinstance(size, Sequence) and len(size) == 1):
        s = size if isinstance(size, int) else size[0]
        w, h = img.size
        if (w <= h and w == s) or (h <= w and h == s):
            return img
        if w < h:
            ow, oh = s, max(1, int(s * h / w))
        else:
            oh, ow = s, max(1, int(s * w / h))
        return img.resize((ow, oh), _PIL_RESAMPLE[interp])
    oh, ow = int(size[0]), int(size[1])
    return img.resize((ow, oh), _PIL_RESAMPLE[interp])


def crop(img, top: int, left: int, height: int, width: int):
    if isinstance(img, torch.Tensor):
        return img[..., top:top + height, left:left + width]
    return img.crop((left, top, left + width, top + height))


def center_crop(img, output_size):
    th, tw = _setup_size2(output_size)
    _, h, w = get_dimensions(img)
    if w < tw or h < th:
        # pad as needed
        pl = max((tw - w) // 2, 0)
        pt = max((th - h) // 2, 0)
        pr = max(tw - w - pl, 0)
        pb = max(th - h - pt, 0)
        img = pad(img, [pl, pt, pr, pb], fill=0)
        _, h, w = get_dimensions(img)
    top = int(round((h - th) / 2.))
    left = int(round((w - tw) / 2.))
    return crop(img, top, left, th, tw)


def _expand_pad(padding):
    if isinstance(padding, numbers.Number):
        return [int(padding)] * 4
    if len(padding) == 1:
        return [int(padding[0])] * 4
    if len(padding) == 2:
        return [int(padding[0]), int(padding[1]), int(padding[0]), int(padding[1])]
    return [int(p) for p in padding]  # l, t, r, b


def pad(img, padding, fill=0, padding_mode='constant'):
    pl, pt, pr, pb = _expand_pad(padding)
    if isinstance(img, torch.Tensor):
        import torch.nn.functional as TF
        mode = {'constant': 'constant', 'reflect': 'reflect', 'edge': 'replicate', 'symmetric': 'reflect'}[padding_mode]
        x = img.unsqueeze(0) if img.ndim == 3 else img
        if mode == 'constant':
            x = TF.pad(x, (pl, pr, pt, pb), mode=mode, value=float(fill if isinstance(fill, (int, float)) else 0))
        else:
            xf = x.float() if not x.is_floating_point() else x
            xf = TF.pad(xf, (pl, pr, pt, pb), mode=mode)
            x = xf.to(x.dtype)
        return x.squeeze(0) if img.ndim == 3 else x
    if padding_mode == 'constant':
        if isinstance(fill, (list, tuple)):
            fill = tuple(int(f) for f in fill)
        return ImageOps.expand(img, border=(pl, pt, pr, pb), fill=fill)
    # reflect/symmetric/edge via numpy
    arr = np.asarray(img)
    np_mode = {'reflect': 'reflect', 'symmetric': 'symmetric', 'edge': 'edge'}[padding_mode]
    if arr.ndim == 3:
        arr = np.pad(arr, ((pt, pb), (pl, pr), (0, 0)), mode=np_mode)
    else:
        arr = np.pad(arr, ((pt, pb), (pl, pr)), mode=np_mode)
    return Image.fromarray(arr)


def resized_crop(img, top, left, height, width, size, interpolation=InterpolationMode.BILINEAR):
    img = crop(img, top, left, height, width)
    return resize(img, size, interpolation)


def hflip(img):
    if isinstance(img, torch.Tensor):
        return img.flip(-1)
    return img.transpose(Image.FLIP_LEFT_RIGHT)


def vflip(img):
    if isinstance(img, torch.Tensor):
        return img.flip(-2)
    return img.transpose(Image.FLIP_TOP_BOTTOM)


def to_tensor(pic) -> torch.Tensor:
    """PIL/ndarray [H,W,C] uint8 -> float tensor [C,H,W] scaled to [0,1]."""
    if isinstance(pic, torch.Tensor):
        return pic
    if isinstance(pic, np.ndarray):
        arr = pic
    else:
        arr = np.array(pic, dtype=np.uint8, copy=True)
    if arr.ndim == 2:
        arr = arr[:, :, None]
    t = torch.from_numpy(arr).permute(2, 0, 1).contiguous()
    return t.float().div_(255.)


def pil_to_tensor(pic) -> torch.Tensor:
    """PIL -> uint8 tensor [C,H,W], no scaling."""
    arr = np.array(pic, copy=True)
    if arr.ndim == 2:
        arr = arr[:, :, None]
    return torch.from_numpy(arr).permute(2, 0, 1).contiguous()


def normalize(tensor: torch.Tensor, mean, std, inplace=False) -> torch.Tensor:
    if not inplace:
        tensor = tensor.clone()
    mean = torch.as_tensor(mean, dtype=tensor.dtype, device=tensor.device)
    std = torch.as_tensor(std, dtype=tensor.dtype, device=tensor.device)
    if mean.ndim == 1:
        mean = mean.view(-1, 1, 1)
    if std.ndim == 1:
        std = std.view(-1, 1, 1)
    return tensor.sub_(mean).div_(std)


def adjust_brightness(img, factor):
    return ImageEnhance.Brightness(img).enhance(factor)


def adjust_contrast(img, factor):
    return ImageEnhance.Contrast(img).enhance(factor)


def adjust_saturation(img, factor):
    return ImageEnhance.Color(img).enhance(factor)


def adjust_hue(img, factor):
    if abs(factor) > 0.5:
        raise ValueError('hue factor out of range')
    if factor == 0:
        return img
    h, s, v = img.convert('HSV').split()
    np_h = np.array(h, dtype=np.uint8)
    np_h = (np_h.astype(np.int16) + int(factor * 255)) % 256
    h = Image.fromarray(np_h.astype(np.uint8), 'L')
    return Image.merge('HSV', (h, s, v)).convert(img.mode)


def rgb_to_grayscale(img, num_output_channels=3):
    g = img.convert('L')
    if num_output_channels == 3:
        return g.convert('RGB')
    return g


def gaussian_blur(img, kernel_size, sigma=None):
    if sigma is None:
        k = kernel_size[0] if isinstance(kernel_size, (list, tuple)) else kernel_size
        sigma = 0.3 * ((k - 1) * 0.5 - 1) + 0.8
    if isinstance(sigma, (list, tuple)):
        sigma = random.uniform(sigma[0], sigma[1])
    return img.filter(ImageFilter.GaussianBlur(radius=sigma))


# ---------------------------------------------------------------------------
# transform classes
# ---------------------------------------------------------------------------

class Compose:
    def __init__(self, transforms):
        self.transforms = list(transforms)

    def __call__(self, img):
        for t in self.transforms:
            img = t(img)
        return img

    def __repr__(self):
        lines = [self.__class__.__name__ + '(']
        for t in self.transforms:
            lines.append(f'    {t}')
        lines.append(')')
        return '\n'.join(lines)


class Resize:
    def __init__(self, size, interpolation=InterpolationMode.BILINEAR, antialias=True):
        self.size = size
        self.interpolation = _to_interp(interpolation)
        self.antialias = antialias

    def __call__(self, img):
        return resize(img, self.size, self.interpolation, antialias=self.antialias)

    def __repr__(self):
        return f'{self.__class__.__name__}(size={self.size}, interpolation={self.interpolation.value})'


class CenterCrop:
    def __init__(self, size):
        self.size = _setup_size2(size)

    def __call__(self, img):
        return center_crop(img, self.size)

    def __repr__(self):
        return f'{self.__class__.__name__}(size={self.size})'


class RandomHorizontalFlip:
    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, img):
        if random.random() < self.p:
            return hflip(img)
        return img

    def __repr__(self):
        return f'{self.__class__.__name__}(p={self.p})'


class RandomVerticalFlip:
    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, img):
        if random.random() < self.p:
            return vflip(img)
        return img

    def __repr__(self):
        return f'{self.__class__.__name__}(p={self.p})'


class ColorJitter:
    """Randomly change brightness/contrast/saturation/hue (PIL)."""

    def __init__(self, brightness=0, contrast=0, saturation=0, hue=0):
        self.brightness = self._check(brightness, 'brightness')
        self.contrast = self._check(contrast, 'contrast')
        self.saturation = self._check(saturation, 'saturation')
        self.hue = self._check(hue, 'hue', center=0, bound=(-0.5, 0.5), clip_first_on_zero=False)

    @staticmethod
    def _check(value, name, center=1, bound=(0, float('inf')), clip_first_on_zero=True):
        if isinstance(value, numbers.Number):
            if value < 0:
                raise ValueError(f'If {name} is a single number, it must be non negative.')
            value = [center - float(value), center + float(value)]
            if clip_first_on_zero:
                value[0] = max(value[0], 0.0)
        elif isinstance(value, (tuple, list)) and len(value) == 2:
            value = [float(value[0]), float(value[1])]
        else:
            raise TypeError(f'{name} should be a single number or a list/tuple with length 2.')
        if value[0] == value[1] == center:
            return None
        return tuple(value)

    def __call__(self, img):
        fns = []
        if self.brightness is not None:
            b = random.uniform(*self.brightness)
            fns.append(lambda im: adjust_brightness(im, b))
        if self.contrast is not None:
            c = random.uniform(*self.contrast)
            fns.append(lambda im: adjust_contrast(im, c))
        if self.saturation is not None:
            s = random.uniform(*self.saturation)
            fns.append(lambda im: adjust_saturation(im, s))
        if self.hue is not None:
            h = random.uniform(*self.hue)
            fns.append(lambda im: adjust_hue(im, h))
        random.shuffle(fns)
        for fn in fns:
            img = fn(img)
        return img

    def __repr__(self):
        return (f'{self.__class__.__name__}(brightness={self.brightness}, contrast={self.contrast}, '
                f'saturation={self.saturation}, hue={self.hue})')


class RandomApply:
    def __init__(self, transforms, p=0.5):
        self.transforms = transforms
        self.p = p

    def __call__(self, img):
        if random.random() < self.p:
            for t in self.transforms:
                img = t(img)
        return img


class RandomGrayscale:
    def __init__(self, p=0.1):
        self.p = p

    def __call__(self, img):
        if random.random() < self.p:
            nc = get_dimensions(img)[0]
            return rgb_to_grayscale(img, num_output_channels=nc)
        return img


class GaussianBlur:
    def __init__(self, kernel_size, sigma=(0.1, 2.0)):
        self.kernel_size = kernel_size
        self.sigma = sigma

    def __call__(self, img):
        return gaussian_blur(img, self.kernel_size, self.sigma)


class Normalize:
    def __init__(self, mean, std, inplace=False):
        self.mean = mean
        self.std = std
        self.inplace = inplace

    def __call__(self, tensor):
        return normalize(tensor, self.mean, self.std, self.inplace)

    def __repr__(self):
        return f'{self.__class__.__name__}(mean={self.mean}, std={self.std})'


class ToTensorTransform:
    def __call__(self, pic):
        return to_tensor(pic)

    def __repr__(self):
        return f'{self.__class__.__name__}()'
