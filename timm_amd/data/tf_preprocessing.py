"""TF-style preprocessing for TF-ported model evaluation (reference
`timm/data/tf_preprocessing.py`, 233 LoC).

TF-trained models (tf_efficientnet, maxvit_*_tf, ...) were evaluated with
TF's crop geometry: a padded center crop of
`size / (size + CROP_PADDING) * min(H, W)` followed by a square resize.
The reference reproduces it by literally running a TF1 graph. This framework
is TF-free: the same geometry (and the TF-train distorted-bbox crop) is
implemented on PIL/numpy, so `--tf-preprocessing` works without installing
TensorFlow. If TF *is* present it can be opted into with
`TIMM_AMD_TF_PREPROCESS=tf` for bit-level TF resize kernels.
"""
import io
import math
import os
import random

import numpy as np
from PIL import Image

IMAGE_SIZE = 224
CROP_PADDING = 32

_PIL_RESAMPLE = {
    'bicubic': Image.BICUBIC,
    'bilinear': Image.BILINEAR,
}


def _decode(image_bytes) -> Image.Image:
    if isinstance(image_bytes, Image.Image):
        return image_bytes.convert('RGB')
    return Image.open(io.BytesIO(image_bytes)).convert('RGB')


def center_crop_tf(img: Image.Image, image_size: int, interpolation: str = 'bicubic') -> Image.Image:
    """TF eval crop: padded center crop then square resize
    (reference `:114-133`)."""
    width, height = img.size
    crop = int((image_size / (image_size + CROP_PADDING)) * min(height, width))
    top = ((height - crop) + 1) // 2
    left = ((width - crop) + 1) // 2
    img = img.crop((left, top, left + crop, top + crop))
    return img.resize((image_size, image_size), _PIL_RESAMPLE[interpolation])


def random_crop_tf(
        img: Image.Image,
        image_size: int,
        interpolation: str = 'bicubic',
        area_range=(0.08, 1.0),
        aspect_ratio_range=(3. / 4, 4. / 3.),
        max_attempts: int = 10,
) -> Image.Image:
    """TF train crop: sampled distorted bounding box, center-crop fallback
    after max_attempts (reference `:33-112`)."""
    width, height = img.size
    area = height * width
    for _ in range(max_attempts):
        target_area = random.uniform(*area_range) * area
        log_ratio = (math.log(aspect_ratio_range[0]), math.log(aspect_ratio_range[1]))
        aspect_ratio = math.exp(random.uniform(*log_ratio))
        w = int(round(math.sqrt(target_area * aspect_ratio)))
        h = int(round(math.sqrt(target_area / aspect_ratio)))
        if 0 < w <= width and 0 < h <= height:
            left = random.randint(0, width - w)
            top = random.randint(0, height - h)
            crop = img.crop((left, top, left + w, top + h))
            return crop.resize((image_size, image_size), _PIL_RESAMPLE[interpolation])
    return center_crop_tf(img, image_size, interpolation)


def preprocess_image(
        image_bytes,
        is_training: bool = False,
        use_bfloat16: bool = False,
        image_size: int = IMAGE_SIZE,
        interpolation: str = 'bicubic',
) -> np.ndarray:
    """Decode + TF-geometry crop/resize -> float HWC array in [0, 255]."""
    img = _decode(image_bytes)
    if is_training:
        img = random_crop_tf(img, image_size, interpolation)
        if random.random() < 0.5:
            img = img.transpose(Image.FLIP_LEFT_RIGHT)
    else:
        img = center_crop_tf(img, image_size, interpolation)
    return np.asarray(img, dtype=np.float32)


class TfPreprocessTransform:
    """Drop-in transform: image bytes (or PIL image) -> CHW uint8 ndarray,
    same output contract as the reference's TF1-session version."""

    def __init__(self, is_training=False, size=224, interpolation='bicubic'):
        self.is_training = is_training
        self.size = size[0] if isinstance(size, tuple) else size
        self.interpolation = interpolation

    def __call__(self, image_bytes):
        img = preprocess_image(
            image_bytes, self.is_training, False, self.size, self.interpolation)
        img = img.round().clip(0, 255).astype(np.uint8)
        if img.ndim < 3:
            img = np.expand_dims(img, axis=-1)
        return np.rollaxis(img, 2)  # HWC -> CHW
