"""Transform-pipeline factory (reference
`timm/data/transforms_factory.py:20,65,273,387`).

Three pipeline builders (no-aug train, full ImageNet train, eval) share one
output-stage rule: with the GPU prefetcher the pipeline ends at a uint8
numpy array (tensor conversion + normalize happen on-device in the fused
prefetch kernel); otherwise it ends in ToTensor+Normalize (or a raw dtype
tensor when normalize=False).
"""
import math
from typing import Optional, Tuple, Union

import torch

from . import image_ops as transforms
from .auto_augment import augment_and_mix_transform, auto_augment_transform, rand_augment_transform
from .constants import DEFAULT_CROP_MODE, DEFAULT_CROP_PCT, IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD
from .random_erasing import RandomErasing
from .transforms import (
    CenterCropOrPad, MaybePILToTensor, MaybeToTensor, RandomCropOrPad,
    RandomResizedCropAndInterpolation, ResizeKeepRatio, ToNumpy, TrimBorder,
    str_to_interp_mode, str_to_pil_interp,
)


def _output_stage(mean, std, use_prefetcher, normalize):
    """Pipeline tail shared by all three builders (see module docstring)."""
    if use_prefetcher:
        return [ToNumpy()]
    if not normalize:
        return [MaybePILToTensor()]
    return [
        MaybeToTensor(),
        transforms.Normalize(mean=torch.tensor(mean), std=torch.tensor(std)),
    ]


def transforms_noaug_train(
        img_size: Union[int, Tuple[int, int]] = 224,
        interpolation: str = 'bilinear',
        mean: Tuple[float, ...] = IMAGENET_DEFAULT_MEAN,
        std: Tuple[float, ...] = IMAGENET_DEFAULT_STD,
        use_prefetcher: bool = False,
        normalize: bool = True,
):
    """Deterministic resize+crop training pipeline (reference `:20`)."""
    if interpolation == 'random':
        interpolation = 'bilinear'  # no random interpolation without aug
    tfl = [
        transforms.Resize(img_size, interpolation=str_to_interp_mode(interpolation)),
        transforms.CenterCrop(img_size),
    ]
    tfl += _output_stage(mean, std, use_prefetcher, normalize)
    return transforms.Compose(tfl)


def _crop_stage(img_size, train_crop_mode, scale, ratio, interpolation):
    """Primary train geometry: RRC default, or resize-keep-ratio + crop/pad."""
    if train_crop_mode in ('rkrc', 'rkrr'):
        scale = tuple(scale or (0.8, 1.00))
        ratio = tuple(ratio or (0.9, 1 / .9))
        crop_cls = CenterCropOrPad if train_crop_mode == 'rkrc' else RandomCropOrPad
        return [
            ResizeKeepRatio(
                img_size,
                interpolation=interpolation,
                random_scale_prob=0.5,
                random_scale_range=scale,
                random_scale_area=True,  # scale semantics compatible with RRC
                random_aspect_prob=0.5,
                random_aspect_range=ratio,
            ),
            crop_cls(img_size, padding_mode='reflect'),
        ]
    scale = tuple(scale or (0.08, 1.0))       # ImageNet defaults
    ratio = tuple(ratio or (3. / 4., 4. / 3.))
    return [RandomResizedCropAndInterpolation(img_size, scale=scale, ratio=ratio, interpolation=interpolation)]


def _auto_augment_stage(auto_augment, img_size, mean, interpolation):
    """Build the AA/RA/AugMix op from a config string."""
    img_size_min = min(img_size) if isinstance(img_size, (tuple, list)) else img_size
    aa_params = dict(
        translate_const=int(img_size_min * 0.45),
        img_mean=tuple(min(255, round(255 * x)) for x in mean),
    )
    if interpolation and interpolation != 'random':
        aa_params['interpolation'] = str_to_pil_interp(interpolation)
    if auto_augment.startswith('rand'):
        return rand_augment_transform(auto_augment, aa_params)
    if auto_augment.startswith('augmix'):
        aa_params['translate_pct'] = 0.3
        return augment_and_mix_transform(auto_augment, aa_params)
    return auto_augment_transform(auto_augment, aa_params)


def transforms_imagenet_train(
        img_size: Union[int, Tuple[int, int]] = 224,
        scale: Optional[Tuple[float, float]] = None,
        ratio: Optional[Tuple[float, float]] = None,
        train_crop_mode: Optional[str] = None,
        hflip: float = 0.5,
        vflip: float = 0.,
        color_jitter: Union[float, Tuple[float, ...]] = 0.4,
        color_jitter_prob: Optional[float] = None,
        force_color_jitter: bool = False,
        grayscale_prob: float = 0.,
        gaussian_blur_prob: float = 0.,
        auto_augment: Optional[str] = None,
        interpolation: str = 'random',
        mean: Tuple[float, ...] = IMAGENET_DEFAULT_MEAN,
        std: Tuple[float, ...] = IMAGENET_DEFAULT_STD,
        re_prob: float = 0.,
        re_mode: str = 'const',
        re_count: int = 1,
        re_num_splits: int = 0,
        use_prefetcher: bool = False,
        normalize: bool = True,
        separate: bool = False,
):
    """Full ImageNet train pipeline (reference `:65`).

    With `separate` the three stages (geometry+flip, color/aug, output+erase)
    come back as a tuple for AugMix split handling.
    """
    train_crop_mode = train_crop_mode or 'rrc'
    assert train_crop_mode in {'rrc', 'rkrc', 'rkrr'}
    primary_tfl = _crop_stage(img_size, train_crop_mode, scale, ratio, interpolation)
    if hflip > 0.:
        primary_tfl += [transforms.RandomHorizontalFlip(p=hflip)]
    if vflip > 0.:
        primary_tfl += [transforms.RandomVerticalFlip(p=vflip)]

    secondary_tfl = []
    disable_color_jitter = False
    if auto_augment:
        assert isinstance(auto_augment, str)
        # AA/RA traditionally replaces color jitter; '3a' policies and the
        # force flag keep both
        disable_color_jitter = not (force_color_jitter or '3a' in auto_augment)
        secondary_tfl += [_auto_augment_stage(auto_augment, img_size, mean, interpolation)]

    if color_jitter is not None and not disable_color_jitter:
        if isinstance(color_jitter, (list, tuple)):
            # 3-tuple = brightness/contrast/saturation, 4th adds hue
            assert len(color_jitter) in (3, 4)
        else:
            color_jitter = (float(color_jitter),) * 3
        jitter = transforms.ColorJitter(*color_jitter)
        if color_jitter_prob is not None:
            jitter = transforms.RandomApply([jitter], p=color_jitter_prob)
        secondary_tfl += [jitter]

    if grayscale_prob:
        secondary_tfl += [transforms.RandomGrayscale(p=grayscale_prob)]
    if gaussian_blur_prob:
        secondary_tfl += [
            transforms.RandomApply([transforms.GaussianBlur(kernel_size=23)], p=gaussian_blur_prob)]

    final_tfl = _output_stage(mean, std, use_prefetcher, normalize)
    if not use_prefetcher and normalize and re_prob > 0.:
        final_tfl += [
            RandomErasing(re_prob, mode=re_mode, max_count=re_count, num_splits=re_num_splits, device='cpu')]

    if separate:
        return (
            transforms.Compose(primary_tfl),
            transforms.Compose(secondary_tfl),
            transforms.Compose(final_tfl),
        )
    return transforms.Compose(primary_tfl + secondary_tfl + final_tfl)


def transforms_imagenet_eval(
        img_size: Union[int, Tuple[int, int]] = 224,
        crop_pct: Optional[float] = None,
        crop_mode: Optional[str] = None,
        crop_border_pixels: Optional[int] = None,
        interpolation: str = 'bilinear',
        mean: Tuple[float, ...] = IMAGENET_DEFAULT_MEAN,
        std: Tuple[float, ...] = IMAGENET_DEFAULT_STD,
        use_prefetcher: bool = False,
        normalize: bool = True,
):
    """Eval pipeline (reference `:273`): resize to img_size/crop_pct then
    crop by mode — 'squash' (no aspect preserve), 'border' (pad to keep all
    pixels) or default center crop."""
    crop_pct = crop_pct or DEFAULT_CROP_PCT

    if isinstance(img_size, (tuple, list)):
        assert len(img_size) == 2
        scale_size = tuple(math.floor(x / crop_pct) for x in img_size)
    else:
        scale_size = (math.floor(img_size / crop_pct),) * 2

    tfl = []
    if crop_border_pixels:
        tfl += [TrimBorder(crop_border_pixels)]

    if crop_mode == 'squash':
        tfl += [
            transforms.Resize(scale_size, interpolation=str_to_interp_mode(interpolation)),
            transforms.CenterCrop(img_size),
        ]
    elif crop_mode == 'border':
        fill = [round(255 * v) for v in mean]
        tfl += [
            ResizeKeepRatio(scale_size, interpolation=interpolation, longest=1.0),
            CenterCropOrPad(img_size, fill=fill),
        ]
    else:
        # center crop: shortest-edge resize keeps aspect, crop discards rest
        if scale_size[0] == scale_size[1]:
            tfl += [transforms.Resize(scale_size[0], interpolation=str_to_interp_mode(interpolation))]
        else:
            tfl += [ResizeKeepRatio(scale_size)]
        tfl += [transforms.CenterCrop(img_size)]

    tfl += _output_stage(mean, std, use_prefetcher, normalize)
    return transforms.Compose(tfl)


def create_transform(
        input_size: Union[int, Tuple[int, int], Tuple[int, int, int]] = 224,
        is_training: bool = False,
        no_aug: bool = False,
        train_crop_mode: Optional[str] = None,
        scale: Optional[Tuple[float, float]] = None,
        ratio: Optional[Tuple[float, float]] = None,
        hflip: float = 0.5,
        vflip: float = 0.,
        color_jitter: Union[float, Tuple[float, ...]] = 0.4,
        color_jitter_prob: Optional[float] = None,
        grayscale_prob: float = 0.,
        gaussian_blur_prob: float = 0.,
        auto_augment: Optional[str] = None,
        interpolation: str = 'bilinear',
        mean: Tuple[float, ...] = IMAGENET_DEFAULT_MEAN,
        std: Tuple[float, ...] = IMAGENET_DEFAULT_STD,
        re_prob: float = 0.,
        re_mode: str = 'const',
        re_count: int = 1,
        re_num_splits: int = 0,
        crop_pct: Optional[float] = None,
        crop_mode: Optional[str] = None,
        crop_border_pixels: Optional[int] = None,
        tf_preprocessing: bool = False,
        use_prefetcher: bool = False,
        normalize: bool = True,
        separate: bool = False,
):
    """Transform dispatcher (reference `:387`)."""
    img_size = input_size[-2:] if isinstance(input_size, (tuple, list)) else input_size

    if tf_preprocessing and use_prefetcher:
        assert not separate, 'Separate transforms not supported for TF preprocessing'
        from .tf_preprocessing import TfPreprocessTransform
        return TfPreprocessTransform(is_training=is_training, size=img_size, interpolation=interpolation)

    if is_training and no_aug:
        assert not separate, 'Cannot perform split augmentation with no_aug'
        return transforms_noaug_train(
            img_size,
            interpolation=interpolation,
            mean=mean,
            std=std,
            use_prefetcher=use_prefetcher,
            normalize=normalize,
        )

    if is_training:
        return transforms_imagenet_train(
            img_size,
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
            re_prob=re_prob,
            re_mode=re_mode,
            re_count=re_count,
            re_num_splits=re_num_splits,
            use_prefetcher=use_prefetcher,
            normalize=normalize,
            separate=separate,
        )

    assert not separate, 'Separate transforms not supported for validation preprocessing'
    return transforms_imagenet_eval(
        img_size,
        interpolation=interpolation,
        mean=mean,
        std=std,
        crop_pct=crop_pct,
        crop_mode=crop_mode,
        crop_border_pixels=crop_border_pixels,
        use_prefetcher=use_prefetcher,
        normalize=normalize,
    )
