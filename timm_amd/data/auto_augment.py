"""AutoAugment / RandAugment / AugMix — string-configured PIL augmentation.

Behavioral parity: /root/reference/timm/data/auto_augment.py (op set,
magnitude scaling laws, policy tables, config-string grammar:
``rand-m9-mstd0.5-inc1``, ``original-mstd0.5``, ``augmix-m5-w4-d2``).

Redesigned around a single op registry: each named op binds its PIL image
function and its magnitude->argument law in one ``OpSpec`` row, rather than
parallel name->fn / name->level-fn dicts of free functions.  Magnitude laws
are inline lambdas over two shared helpers (``frac`` = m/10 fractional level,
``sgn`` = random sign flip).  All ops run host-side in loader workers.
"""
import math
import random
import re
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Union

import numpy as np
import PIL
from PIL import Image, ImageEnhance, ImageFilter, ImageOps

_PIL_VER = tuple(int(x) for x in PIL.__version__.split('.')[:2])

MAX_MAG = 10.  # the 'M' scale: magnitudes are fractions of this denominator

_FILL = (128, 128, 128)

_HPARAMS_DEFAULT = dict(
    translate_const=250,
    img_mean=_FILL,
)

if hasattr(Image, 'Resampling'):
    _RANDOM_INTERPOLATION = (Image.Resampling.BILINEAR, Image.Resampling.BICUBIC)
    _DEFAULT_INTERPOLATION = Image.Resampling.BICUBIC
else:
    _RANDOM_INTERPOLATION = (Image.BILINEAR, Image.BICUBIC)
    _DEFAULT_INTERPOLATION = Image.BICUBIC


# ---------------------------------------------------------------------------
# image ops (PIL)
# ---------------------------------------------------------------------------

def _geom_kwargs(kwargs):
    """Resolve resample (random choice among tuple) + drop unsupported fill."""
    resample = kwargs.pop('resample', _DEFAULT_INTERPOLATION)
    if isinstance(resample, (list, tuple)):
        resample = random.choice(resample)
    kwargs['resample'] = resample
    if _PIL_VER < (5, 0):
        kwargs.pop('fillcolor', None)
    return kwargs


def _affine(img, coeffs, **kwargs):
    return img.transform(img.size, Image.AFFINE, coeffs, **_geom_kwargs(kwargs))


def _shear_x(img, factor, **kw):
    return _affine(img, (1, factor, 0, 0, 1, 0), **kw)


def _shear_y(img, factor, **kw):
    return _affine(img, (1, 0, 0, factor, 1, 0), **kw)


def _translate_x_abs(img, pixels, **kw):
    return _affine(img, (1, 0, pixels, 0, 1, 0), **kw)


def _translate_y_abs(img, pixels, **kw):
    return _affine(img, (1, 0, 0, 0, 1, pixels), **kw)


def _translate_x_rel(img, pct, **kw):
    return _translate_x_abs(img, pct * img.size[0], **kw)


def _translate_y_rel(img, pct, **kw):
    return _translate_y_abs(img, pct * img.size[1], **kw)


def _rotate(img, degrees, **kw):
    kw = _geom_kwargs(kw)
    if _PIL_VER >= (5, 2):
        return img.rotate(degrees, **kw)
    if _PIL_VER >= (5, 0):
        # manual center-rotation affine for PIL 5.0/5.1 (no fill in rotate)
        w, h = img.size
        cx, cy = w / 2.0, h / 2.0
        angle = -math.radians(degrees)
        cos_a = round(math.cos(angle), 15)
        sin_a = round(math.sin(angle), 15)
        tx = cos_a * -cx + sin_a * -cy + cx
        ty = -sin_a * -cx + cos_a * -cy + cy
        return img.transform(img.size, Image.AFFINE, (cos_a, sin_a, tx, -sin_a, cos_a, ty), **kw)
    return img.rotate(degrees, resample=kw['resample'])


def _solarize_add(img, add, thresh=128, **__):
    # lift values below thresh by `add`, clamped to 255
    lut = [min(255, i + add) if i < thresh else i for i in range(256)]
    if img.mode in ('L', 'RGB'):
        return img.point(lut * 3 if img.mode == 'RGB' else lut)
    return img


def _posterize(img, bits_to_keep, **__):
    return img if bits_to_keep >= 8 else ImageOps.posterize(img, bits_to_keep)


def _gaussian_blur(img, radius, **__):
    return img.filter(ImageFilter.GaussianBlur(radius=radius))


def _gaussian_blur_rand(img, factor, **__):
    # radius uniform in [0.1, 2*factor]
    return img.filter(ImageFilter.GaussianBlur(
        radius=random.uniform(0.1, 2.0 * factor)))


def _desaturate(img, factor, **__):
    # factor 1 -> grayscale, 0 -> unchanged (enhance arg inverted+clamped)
    return ImageEnhance.Color(img).enhance(min(1., max(0., 1. - factor)))


def _enhance(enhancer):
    return lambda img, factor, **__: enhancer(img).enhance(factor)


# ---------------------------------------------------------------------------
# magnitude laws
# ---------------------------------------------------------------------------

def frac(m):
    """Magnitude as a fraction of the M-scale denominator."""
    return m / MAX_MAG


def sgn(v):
    """Randomly flip sign with p=0.5."""
    return -v if random.random() > 0.5 else v


def _span(m, lo, hi, clamp=True):
    v = lo + (hi - lo) * frac(m)
    return max(lo, min(hi, v)) if clamp else v


def _enh(m, _hp):
    # blend factor in [0.1, 1.9], severity symmetric around 1.0? no: linear
    return (frac(m) * 1.8 + 0.1,)


def _enh_inc(m, _hp):
    # 'no change' is 1.0; severity grows away from it in a random direction
    return (max(0.1, 1.0 + sgn(frac(m) * .9)),)


@dataclass(frozen=True)
class OpSpec:
    """One augmentation: PIL fn + magnitude->args law (None = no args)."""
    fn: Callable
    args: Optional[Callable] = None


AUG_OPS: Dict[str, OpSpec] = {
    'AutoContrast': OpSpec(lambda img, **__: ImageOps.autocontrast(img)),
    'Equalize': OpSpec(lambda img, **__: ImageOps.equalize(img)),
    'Invert': OpSpec(lambda img, **__: ImageOps.invert(img)),
    'Rotate': OpSpec(_rotate, lambda m, hp: (sgn(frac(m) * 30.),)),
    'ShearX': OpSpec(_shear_x, lambda m, hp: (sgn(frac(m) * 0.3),)),
    'ShearY': OpSpec(_shear_y, lambda m, hp: (sgn(frac(m) * 0.3),)),
    'TranslateX': OpSpec(
        _translate_x_abs, lambda m, hp: (sgn(frac(m) * float(hp['translate_const'])),)),
    'TranslateY': OpSpec(
        _translate_y_abs, lambda m, hp: (sgn(frac(m) * float(hp['translate_const'])),)),
    'TranslateXRel': OpSpec(
        _translate_x_rel, lambda m, hp: (sgn(frac(m) * hp.get('translate_pct', 0.45)),)),
    'TranslateYRel': OpSpec(
        _translate_y_rel, lambda m, hp: (sgn(frac(m) * hp.get('translate_pct', 0.45)),)),
    # posterize family: bits kept; plain = severity falls with m, increasing =
    # severity rises, original = AutoAugment-paper 4..8 bit range
    'Posterize': OpSpec(_posterize, lambda m, hp: (int(frac(m) * 4),)),
    'PosterizeIncreasing': OpSpec(_posterize, lambda m, hp: (4 - int(frac(m) * 4),)),
    'PosterizeOriginal': OpSpec(_posterize, lambda m, hp: (int(frac(m) * 4) + 4,)),
    'Solarize': OpSpec(
        lambda img, t, **__: ImageOps.solarize(img, t),
        lambda m, hp: (min(256, int(frac(m) * 256)),)),
    'SolarizeIncreasing': OpSpec(
        lambda img, t, **__: ImageOps.solarize(img, t),
        lambda m, hp: (256 - min(256, int(frac(m) * 256)),)),
    'SolarizeAdd': OpSpec(_solarize_add, lambda m, hp: (min(128, int(frac(m) * 110)),)),
    'Color': OpSpec(_enhance(ImageEnhance.Color), _enh),
    'ColorIncreasing': OpSpec(_enhance(ImageEnhance.Color), _enh_inc),
    'Contrast': OpSpec(_enhance(ImageEnhance.Contrast), _enh),
    'ContrastIncreasing': OpSpec(_enhance(ImageEnhance.Contrast), _enh_inc),
    'Brightness': OpSpec(_enhance(ImageEnhance.Brightness), _enh),
    'BrightnessIncreasing': OpSpec(_enhance(ImageEnhance.Brightness), _enh_inc),
    'Sharpness': OpSpec(_enhance(ImageEnhance.Sharpness), _enh),
    'SharpnessIncreasing': OpSpec(_enhance(ImageEnhance.Sharpness), _enh_inc),
    'Desaturate': OpSpec(_desaturate, lambda m, hp: (_span(m, 0.5, 1.0),)),
    'GaussianBlur': OpSpec(_gaussian_blur, lambda m, hp: (_span(m, 0.1, 2.0),)),
    'GaussianBlurRand': OpSpec(_gaussian_blur_rand, lambda m, hp: (_span(m, 0., 1.0),)),
}

# reference-compatible aliases for external pokes
NAME_TO_OP = {name: spec.fn for name, spec in AUG_OPS.items()}
LEVEL_TO_ARG = {name: spec.args for name, spec in AUG_OPS.items()}


class AugmentOp:
    """A registered op bound to (probability, magnitude, hparams).

    magnitude_std hparam > 0 gausses the magnitude per call (inf = uniform in
    [0, m]); magnitude_max overrides the [0, 10] clamp ceiling.
    """

    def __init__(self, name, prob=0.5, magnitude=10, hparams=None):
        hparams = hparams or _HPARAMS_DEFAULT
        self.name = name
        spec = AUG_OPS[name]
        self.aug_fn = spec.fn
        self.level_fn = spec.args
        self.prob = prob
        self.magnitude = magnitude
        self.hparams = hparams.copy()
        self.kwargs = dict(
            fillcolor=hparams.get('img_mean', _FILL),
            resample=hparams.get('interpolation', _RANDOM_INTERPOLATION),
        )
        self.magnitude_std = self.hparams.get('magnitude_std', 0)
        self.magnitude_max = self.hparams.get('magnitude_max', None)

    def _sample_magnitude(self):
        m = self.magnitude
        if self.magnitude_std > 0:
            if self.magnitude_std == float('inf'):
                m = random.uniform(0, m)
            else:
                m = random.gauss(m, self.magnitude_std)
        ceil = self.magnitude_max or MAX_MAG
        return max(0., min(m, ceil))

    def __call__(self, img):
        if self.prob < 1.0 and random.random() > self.prob:
            return img
        # magnitude is sampled even for no-arg ops so the RNG stream matches
        # runs with any op mix (keeps aug reproducibility format-stable)
        magnitude = self._sample_magnitude()
        args = self.level_fn(magnitude, self.hparams) if self.level_fn is not None else ()
        return self.aug_fn(img, *args, **self.kwargs)

    def __repr__(self):
        s = f'{self.__class__.__name__}(name={self.name}, p={self.prob}'
        s += f', m={self.magnitude}, mstd={self.magnitude_std}'
        if self.magnitude_max is not None:
            s += f', mmax={self.magnitude_max}'
        return s + ')'


# ---------------------------------------------------------------------------
# config-string parsing (shared grammar: dash-separated "key<number>" tokens)
# ---------------------------------------------------------------------------

def _iter_config(tokens):
    """Yield (key, raw_value) for each 'key123'-style token; skip bare keys."""
    for tok in tokens:
        parts = re.split(r'(\d.*)', tok)
        if len(parts) >= 2:
            yield parts[0], parts[1]


# ---------------------------------------------------------------------------
# AutoAugment (fixed sub-policy tables)
# ---------------------------------------------------------------------------

# (name, prob, magnitude) pairs; applied in sequence when the sub-policy is
# drawn.  Tables transcribed from the TF TPU EfficientNet / AutoAugment-paper
# policies (data constants, shared with the reference).
_POLICY_V0 = [
    [('Equalize', 0.8, 1), ('ShearY', 0.8, 4)],
    [('Color', 0.4, 9), ('Equalize', 0.6, 3)],
    [('Color', 0.4, 1), ('Rotate', 0.6, 8)],
    [('Solarize', 0.8, 3), ('Equalize', 0.4, 7)],
    [('Solarize', 0.4, 2), ('Solarize', 0.6, 2)],
    [('Color', 0.2, 0), ('Equalize', 0.8, 8)],
    [('Equalize', 0.4, 8), ('SolarizeAdd', 0.8, 3)],
    [('ShearX', 0.2, 9), ('Rotate', 0.6, 8)],
    [('Color', 0.6, 1), ('Equalize', 1.0, 2)],
    [('Invert', 0.4, 9), ('Rotate', 0.6, 0)],
    [('Equalize', 1.0, 9), ('ShearY', 0.6, 3)],
    [('Color', 0.4, 7), ('Equalize', 0.6, 0)],
    [('Posterize', 0.4, 6), ('AutoContrast', 0.4, 7)],
    [('Solarize', 0.6, 8), ('Color', 0.6, 9)],
    [('Solarize', 0.2, 4), ('Rotate', 0.8, 9)],
    [('Rotate', 1.0, 7), ('TranslateYRel', 0.8, 9)],
    [('ShearX', 0.0, 0), ('Solarize', 0.8, 4)],
    [('ShearY', 0.8, 0), ('Color', 0.6, 4)],
    [('Color', 1.0, 0), ('Rotate', 0.6, 2)],
    [('Equalize', 0.8, 4), ('Equalize', 0.0, 8)],
    [('Equalize', 1.0, 4), ('AutoContrast', 0.6, 2)],
    [('ShearY', 0.4, 7), ('SolarizeAdd', 0.6, 7)],
    [('Posterize', 0.8, 2), ('Solarize', 0.6, 10)],
    [('Solarize', 0.6, 8), ('Equalize', 0.6, 1)],
    [('Color', 0.8, 6), ('Rotate', 0.4, 5)],
]

_POLICY_ORIGINAL = [
    [('PosterizeOriginal', 0.4, 8), ('Rotate', 0.6, 9)],
    [('Solarize', 0.6, 5), ('AutoContrast', 0.6, 5)],
    [('Equalize', 0.8, 8), ('Equalize', 0.6, 3)],
    [('PosterizeOriginal', 0.6, 7), ('PosterizeOriginal', 0.6, 6)],
    [('Equalize', 0.4, 7), ('Solarize', 0.2, 4)],
    [('Equalize', 0.4, 4), ('Rotate', 0.8, 8)],
    [('Solarize', 0.6, 3), ('Equalize', 0.6, 7)],
    [('PosterizeOriginal', 0.8, 5), ('Equalize', 1.0, 2)],
    [('Rotate', 0.2, 3), ('Solarize', 0.6, 8)],
    [('Equalize', 0.6, 8), ('PosterizeOriginal', 0.4, 6)],
    [('Rotate', 0.8, 8), ('Color', 0.4, 0)],
    [('Rotate', 0.4, 9), ('Equalize', 0.6, 2)],
    [('Equalize', 0.0, 7), ('Equalize', 0.8, 8)],
    [('Invert', 0.6, 4), ('Equalize', 1.0, 8)],
    [('Color', 0.6, 4), ('Contrast', 1.0, 8)],
    [('Rotate', 0.8, 8), ('Color', 1.0, 2)],
    [('Color', 0.8, 8), ('Solarize', 0.8, 7)],
    [('Sharpness', 0.4, 7), ('Invert', 0.6, 8)],
    [('ShearX', 0.6, 5), ('Equalize', 1.0, 9)],
    [('Color', 0.4, 0), ('Equalize', 0.6, 3)],
    [('Equalize', 0.4, 7), ('Solarize', 0.2, 4)],
    [('Solarize', 0.6, 5), ('AutoContrast', 0.6, 5)],
    [('Invert', 0.6, 4), ('Equalize', 1.0, 8)],
    [('Color', 0.6, 4), ('Contrast', 1.0, 8)],
    [('Equalize', 0.8, 8), ('Equalize', 0.6, 3)],
]

_POLICY_3A = [
    [('Solarize', 1.0, 5)],
    [('Desaturate', 1.0, 10)],
    [('GaussianBlurRand', 1.0, 10)],
]


def _swap_increasing(table):
    """'r' policy variants: posterize ops become increasing-severity."""
    repl = {
        'Posterize': 'PosterizeIncreasing',
        'PosterizeOriginal': 'PosterizeIncreasing',
    }
    return [[(repl.get(n, n), p, m) for n, p, m in sub] for sub in table]


_POLICIES = {
    'v0': _POLICY_V0,
    'v0r': _swap_increasing(_POLICY_V0),
    'original': _POLICY_ORIGINAL,
    'originalr': _swap_increasing(_POLICY_ORIGINAL),
    '3a': _POLICY_3A,
}


def auto_augment_policy(name='v0', hparams=None):
    hparams = hparams or _HPARAMS_DEFAULT
    assert name in _POLICIES, f'Unknown AA policy {name}'
    return [
        [AugmentOp(*args, hparams=hparams) for args in sub]
        for sub in _POLICIES[name]
    ]


# per-name policy fns kept for reference-API compat
def auto_augment_policy_v0(hparams):
    return auto_augment_policy('v0', hparams)


def auto_augment_policy_v0r(hparams):
    return auto_augment_policy('v0r', hparams)


def auto_augment_policy_original(hparams):
    return auto_augment_policy('original', hparams)


def auto_augment_policy_originalr(hparams):
    return auto_augment_policy('originalr', hparams)


def auto_augment_policy_3a(hparams):
    return auto_augment_policy('3a', hparams)


class AutoAugment:
    """Draw one sub-policy per image and apply its ops in order."""

    def __init__(self, policy):
        self.policy = policy

    def __call__(self, img):
        for op in random.choice(self.policy):
            img = op(img)
        return img

    def __repr__(self):
        body = ''.join(
            '\n\t[' + ', '.join(str(op) for op in sub) + ']' for sub in self.policy)
        return f'{self.__class__.__name__}(policy={body})'


def auto_augment_transform(config_str: str, hparams: Optional[Dict] = None):
    """Build AutoAugment from e.g. 'original-mstd0.5' (policy name + hparams)."""
    name, *rest = config_str.split('-')
    for key, val in _iter_config(rest):
        assert key == 'mstd', 'Unknown AutoAugment config section'
        hparams.setdefault('magnitude_std', float(val))
    return AutoAugment(auto_augment_policy(name, hparams=hparams))


# ---------------------------------------------------------------------------
# RandAugment
# ---------------------------------------------------------------------------

_RAND_TRANSFORMS = [
    'AutoContrast', 'Equalize', 'Invert', 'Rotate',
    'Posterize', 'Solarize', 'SolarizeAdd',
    'Color', 'Contrast', 'Brightness', 'Sharpness',
    'ShearX', 'ShearY', 'TranslateXRel', 'TranslateYRel',
]

_RAND_INCREASING_TRANSFORMS = [
    'AutoContrast', 'Equalize', 'Invert', 'Rotate',
    'PosterizeIncreasing', 'SolarizeIncreasing', 'SolarizeAdd',
    'ColorIncreasing', 'ContrastIncreasing', 'BrightnessIncreasing',
    'SharpnessIncreasing', 'ShearX', 'ShearY', 'TranslateXRel', 'TranslateYRel',
]

_RAND_3A = ['SolarizeIncreasing', 'Desaturate', 'GaussianBlur']

_RAND_WEIGHTED_3A = {
    'SolarizeIncreasing': 6, 'Desaturate': 6, 'GaussianBlur': 6,
    'Rotate': 3, 'ShearX': 2, 'ShearY': 2,
    'PosterizeIncreasing': 1, 'AutoContrast': 1, 'ColorIncreasing': 1,
    'SharpnessIncreasing': 1, 'ContrastIncreasing': 1, 'BrightnessIncreasing': 1,
    'Equalize': 1, 'Invert': 1,
}

_RAND_WEIGHTED_0 = {
    'Rotate': 3, 'ShearX': 2, 'ShearY': 2,
    'TranslateXRel': 1, 'TranslateYRel': 1,
    'ColorIncreasing': .25, 'SharpnessIncreasing': 0.25, 'AutoContrast': 0.25,
    'SolarizeIncreasing': .05, 'SolarizeAdd': .05, 'ContrastIncreasing': .05,
    'BrightnessIncreasing': .05, 'Equalize': .05, 'PosterizeIncreasing': 0.05,
    'Invert': .05,
}

_RAND_CHOICE_SETS = {
    'weights': _RAND_WEIGHTED_0,
    '3aw': _RAND_WEIGHTED_3A,
    '3a': _RAND_3A,
}


def rand_augment_choices(name: str, increasing=True):
    if name in _RAND_CHOICE_SETS:
        return _RAND_CHOICE_SETS[name]
    return _RAND_INCREASING_TRANSFORMS if increasing else _RAND_TRANSFORMS


def _normalize_weights(weighted: Dict):
    names, weights = zip(*weighted.items())
    weights = np.array(weights)
    return names, weights / weights.sum()


def rand_augment_ops(
        magnitude: Union[int, float] = 10,
        prob: float = 0.5,
        hparams: Optional[Dict] = None,
        transforms: Optional[Union[Dict, List]] = None,
):
    hparams = hparams or _HPARAMS_DEFAULT
    transforms = transforms or _RAND_TRANSFORMS
    return [
        AugmentOp(name, prob=prob, magnitude=magnitude, hparams=hparams)
        for name in transforms
    ]


class RandAugment:
    """Apply num_layers ops drawn from the op pool (weighted = no-replace)."""

    def __init__(self, ops, num_layers=2, choice_weights=None):
        self.ops = ops
        self.num_layers = num_layers
        self.choice_weights = choice_weights

    def __call__(self, img):
        chosen = np.random.choice(
            self.ops, self.num_layers,
            replace=self.choice_weights is None,
            p=self.choice_weights,
        )
        for op in chosen:
            img = op(img)
        return img

    def __repr__(self):
        ops = ''.join(f'\n\t{op}' for op in self.ops)
        return f'{self.__class__.__name__}(n={self.num_layers}, ops={ops})'


def rand_augment_transform(
        config_str: str,
        hparams: Optional[Dict] = None,
        transforms: Optional[Union[str, Dict, List]] = None,
):
    """Build RandAugment from e.g. 'rand-m9-n3-mstd0.5'.

    Keys: m magnitude · n layers · p per-op prob · mstd magnitude noise
    (>100 = uniform) · mmax magnitude ceiling · inc increasing-severity sets ·
    t<name> named transform set.
    """
    tokens = config_str.split('-')
    assert tokens[0] == 'rand'
    magnitude, num_layers, prob, increasing = MAX_MAG, 2, 0.5, False
    for tok in tokens[1:]:
        if tok.startswith('t'):
            if transforms is None:
                transforms = str(tok[1:])
            continue
        for key, val in _iter_config([tok]):
            if key == 'm':
                magnitude = int(val)
            elif key == 'n':
                num_layers = int(val)
            elif key == 'p':
                prob = float(val)
            elif key == 'mstd':
                mstd = float(val)
                hparams.setdefault(
                    'magnitude_std', float('inf') if mstd > 100 else mstd)
            elif key == 'mmax':
                hparams.setdefault('magnitude_max', int(val))
            elif key == 'inc':
                increasing = increasing or bool(val)
            else:
                assert False, 'Unknown RandAugment config section'

    if isinstance(transforms, str):
        transforms = rand_augment_choices(transforms, increasing=increasing)
    elif transforms is None:
        transforms = _RAND_INCREASING_TRANSFORMS if increasing else _RAND_TRANSFORMS
    choice_weights = None
    if isinstance(transforms, Dict):
        transforms, choice_weights = _normalize_weights(transforms)
    ops = rand_augment_ops(
        magnitude=magnitude, prob=prob, hparams=hparams, transforms=transforms)
    return RandAugment(ops, num_layers, choice_weights=choice_weights)


# ---------------------------------------------------------------------------
# AugMix
# ---------------------------------------------------------------------------

_AUGMIX_TRANSFORMS = [
    'AutoContrast', 'ColorIncreasing', 'ContrastIncreasing',
    'BrightnessIncreasing', 'SharpnessIncreasing', 'Equalize',
    'Rotate', 'PosterizeIncreasing', 'SolarizeIncreasing',
    'ShearX', 'ShearY', 'TranslateXRel', 'TranslateYRel',
]


def augmix_ops(
        magnitude: Union[int, float] = 10,
        hparams: Optional[Dict] = None,
        transforms: Optional[Union[str, Dict, List]] = None,
):
    hparams = hparams or _HPARAMS_DEFAULT
    transforms = transforms or _AUGMIX_TRANSFORMS
    return [
        AugmentOp(name, prob=1.0, magnitude=magnitude, hparams=hparams)
        for name in transforms
    ]


class AugMixAugment:
    """AugMix (arxiv 1912.02781): blend `width` random op chains with
    Dirichlet weights, then blend with the original by a Beta draw."""

    def __init__(self, ops, alpha=1., width=3, depth=-1, blended=False):
        self.ops = ops
        self.alpha = alpha
        self.width = width
        self.depth = depth
        self.blended = blended  # sequential PIL blends instead of numpy accum

    def _chain(self, img):
        depth = self.depth if self.depth > 0 else np.random.randint(1, 4)
        for op in np.random.choice(self.ops, depth, replace=True):
            img = op(img)
        return img

    def _apply_blended(self, img, mixing_weights, m):
        # sequential pairwise blends equivalent to the weighted sum: chain i
        # gets alpha_i = w_i / remaining mass
        ws = mixing_weights * m
        cumulative = 1.
        alphas = []
        for w in ws[::-1]:
            alphas.append(w / cumulative)
            cumulative *= 1 - alphas[-1]
        original = img.copy()
        for a in reversed(alphas):
            img = Image.blend(img, self._chain(original), a)
        return img

    def _apply_basic(self, img, mixing_weights, m):
        # note: numpy image arrays are (H, W, C); the reference builds the
        # accumulator (W, H, C) and crashes on non-square inputs
        shape = img.size[1], img.size[0], len(img.getbands())
        accum = np.zeros(shape, dtype=np.float32)
        for w in mixing_weights:
            accum += w * np.asarray(self._chain(img), dtype=np.float32)
        np.clip(accum, 0, 255., out=accum)
        return Image.blend(img, Image.fromarray(accum.astype(np.uint8)), m)

    def __call__(self, img):
        mixing_weights = np.float32(np.random.dirichlet([self.alpha] * self.width))
        m = np.float32(np.random.beta(self.alpha, self.alpha))
        apply = self._apply_blended if self.blended else self._apply_basic
        return apply(img, mixing_weights, m)

    def __repr__(self):
        ops = ''.join(f'\n\t{op}' for op in self.ops)
        return (f'{self.__class__.__name__}(alpha={self.alpha}, w={self.width}, '
                f'd={self.depth}, ops={ops})')


def augment_and_mix_transform(config_str: str, hparams: Optional[Dict] = None):
    """Build AugMix from e.g. 'augmix-m5-w4-d2'.

    Keys: m magnitude · w width · d depth · a alpha · b blended-mode ·
    mstd magnitude noise.  Magnitude sampling defaults to uniform [0, m].
    """
    tokens = config_str.split('-')
    assert tokens[0] == 'augmix'
    params = dict(magnitude=3, width=3, depth=-1, alpha=1., blended=False)
    for key, val in _iter_config(tokens[1:]):
        if key == 'mstd':
            hparams.setdefault('magnitude_std', float(val))
        elif key == 'm':
            params['magnitude'] = int(val)
        elif key == 'w':
            params['width'] = int(val)
        elif key == 'd':
            params['depth'] = int(val)
        elif key == 'a':
            params['alpha'] = float(val)
        elif key == 'b':
            params['blended'] = bool(val)
        else:
            assert False, 'Unknown AugMix config section'
    hparams.setdefault('magnitude_std', float('inf'))
    ops = augmix_ops(magnitude=params.pop('magnitude'), hparams=hparams)
    return AugMixAugment(ops, **params)
