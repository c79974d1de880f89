"""Variable-size Mixup/CutMix for the NaFlex pipeline.

Behavioral parity: /root/reference/timm/data/naflex_mixup.py (central-overlap
mixing of aspect-paired variable-size images, provenance-exact soft targets).

Images in a NaFlex batch have different (H, W); resizing for mixing would
defeat the aspect-preserving pipeline, so each aspect-matched pair mixes only
its mutual central crop and the per-sample lambda is corrected to the true
fraction of own pixels.
"""
import math
import random
from typing import Dict, List, Tuple

import torch

__all__ = [
    'mix_batch_variable_size', 'pairwise_mixup_target', 'smoothed_sparse_target',
    'NaFlexMixup',
]


def _draw_mode(mixup_alpha: float, cutmix_alpha: float, switch_prob: float):
    """Pick (use_cutmix, lambda) for this batch."""
    if mixup_alpha > 0.0 and cutmix_alpha > 0.0:
        use_cutmix = torch.rand(()).item() < switch_prob
        alpha = cutmix_alpha if use_cutmix else mixup_alpha
    elif mixup_alpha > 0.0:
        use_cutmix, alpha = False, mixup_alpha
    elif cutmix_alpha > 0.0:
        use_cutmix, alpha = True, cutmix_alpha
    else:
        raise ValueError('Both mixup_alpha and cutmix_alpha are zero – nothing to do.')
    lam = torch.distributions.Beta(alpha, alpha).sample().item()
    return use_cutmix, min(1.0, max(0.0, lam))


def _aspect_pairs(imgs: List[torch.Tensor], local_shuffle: int):
    """Pair neighbours after an aspect-ratio sort (+ windowed shuffle so the
    pairings vary across epochs).  Returns (pair map, leftover index)."""
    order = sorted(range(len(imgs)), key=lambda i: imgs[i].shape[2] / imgs[i].shape[1])
    if local_shuffle > 1:
        for start in range(0, len(order), local_shuffle):
            random.shuffle(order[start:start + local_shuffle])
    pair_to: Dict[int, int] = {}
    for a, b in zip(order[::2], order[1::2]):
        pair_to[a], pair_to[b] = b, a
    leftover = order[-1] if len(imgs) % 2 else None
    return pair_to, leftover


def _overlap_window(xi, xj):
    """Central region common to both images: (oh, ow, (top_i, left_i), (top_j, left_j))."""
    _, hi, wi = xi.shape
    _, hj, wj = xj.shape
    oh, ow = min(hi, hj), min(wi, wj)
    return oh, ow, ((hi - oh) // 2, (wi - ow) // 2), ((hj - oh) // 2, (wj - ow) // 2)


def mix_batch_variable_size(
        imgs: List[torch.Tensor],
        *,
        mixup_alpha: float = 0.8,
        cutmix_alpha: float = 1.0,
        switch_prob: float = 0.5,
        local_shuffle: int = 4,
) -> Tuple[List[torch.Tensor], List[float], Dict[int, int]]:
    """Mixup/CutMix over a list of (C, H, W) images of varying size.

    Returns (mixed images, per-sample lambda = fraction of own pixels,
    pairing map i -> j).
    """
    if len(imgs) < 2:
        raise ValueError('Need at least two images to perform Mixup/CutMix.')
    use_cutmix, lam_raw = _draw_mode(mixup_alpha, cutmix_alpha, switch_prob)
    pair_to, leftover = _aspect_pairs(imgs, local_shuffle)

    mixed: List[torch.Tensor] = [None] * len(imgs)
    lams: List[float] = [1.0] * len(imgs)

    for i, xi in enumerate(imgs):
        if i == leftover:
            mixed[i] = xi
            continue
        xj = imgs[pair_to[i]]
        oh, ow, (ti, li), (tj, lj) = _overlap_window(xi, xj)
        own_area = xi.shape[1] * xi.shape[2]
        out = xi.clone()

        if use_cutmix:
            # random rectangle inside the shared window
            side = math.sqrt(1.0 - lam_raw)
            ch, cw = int(oh * side), int(ow * side)
            dy = random.randint(0, oh - ch)
            dx = random.randint(0, ow - cw)
            out[:, ti + dy:ti + dy + ch, li + dx:li + dx + cw] = \
                xj[:, tj + dy:tj + dy + ch, lj + dx:lj + dx + cw]
            lams[i] = 1.0 - (ch * cw) / float(own_area)
        else:
            # blend the whole shared window; pixels outside it stay 'own'
            own = out[:, ti:ti + oh, li:li + ow]
            other = xj[:, tj:tj + oh, lj:lj + ow]
            out[:, ti:ti + oh, li:li + ow] = own.mul(lam_raw).add_(other, alpha=1.0 - lam_raw)
            overlap = oh * ow
            lams[i] = (own_area - overlap) / own_area + lam_raw * overlap / own_area
        mixed[i] = out

    return mixed, lams, pair_to


def smoothed_sparse_target(
        targets: torch.Tensor,
        *,
        num_classes: int,
        smoothing: float = 0.0,
) -> torch.Tensor:
    low = smoothing / num_classes
    high = 1.0 - smoothing + low
    dense = torch.full(
        (targets.size(0), num_classes), low, dtype=torch.float32, device=targets.device)
    return dense.scatter_(1, targets.unsqueeze(1), high)


def pairwise_mixup_target(
        targets: torch.Tensor,
        pair_to: Dict[int, int],
        lam_list: List[float],
        *,
        num_classes: int,
        smoothing: float = 0.0,
) -> torch.Tensor:
    """Soft targets matching the exact pixel provenance of the mixer."""
    base = smoothed_sparse_target(targets, num_classes=num_classes, smoothing=smoothing)
    out = base.clone()
    for i, j in pair_to.items():
        out[i].mul_(lam_list[i]).add_(base[j], alpha=1.0 - lam_list[i])
    return out


class NaFlexMixup:
    """Functor bundling the variable-size mixer + target builder."""

    def __init__(
            self,
            *,
            num_classes: int,
            mixup_alpha: float = 0.8,
            cutmix_alpha: float = 1.0,
            switch_prob: float = 0.5,
            prob: float = 1.0,
            local_shuffle: int = 4,
            label_smoothing: float = 0.0,
    ) -> None:
        self.num_classes = num_classes
        self.mixup_alpha = mixup_alpha
        self.cutmix_alpha = cutmix_alpha
        self.switch_prob = switch_prob
        self.prob = prob
        self.local_shuffle = local_shuffle
        self.smoothing = label_smoothing

    def __call__(self, imgs: List[torch.Tensor], targets: torch.Tensor):
        if not isinstance(targets, torch.Tensor):
            targets = torch.tensor(targets)
        if random.random() > self.prob:
            soft = smoothed_sparse_target(
                targets, num_classes=self.num_classes, smoothing=self.smoothing)
            return imgs, soft.unbind(0)
        mixed, lams, pair_to = mix_batch_variable_size(
            imgs,
            mixup_alpha=self.mixup_alpha,
            cutmix_alpha=self.cutmix_alpha,
            switch_prob=self.switch_prob,
            local_shuffle=self.local_shuffle,
        )
        soft = pairwise_mixup_target(
            targets, pair_to, lams,
            num_classes=self.num_classes, smoothing=self.smoothing)
        return mixed, soft.unbind(0)
