"""Mixup and CutMix batch augmentation.

Behavioral parity: /root/reference/timm/data/mixup.py:90,221 (batch/pair/elem
/half modes, lambda correction, uint8 collate-time mixing, soft targets).

Redesign notes: one `_MixParams` sampler produces (lambda, use_cutmix) draws
for every mode; the element/pair loops share a single `_cut_coords` helper.
`Mixup` operates in-place on device tensors after collate; `FastCollateMixup`
performs the same math on uint8 numpy frames inside the collate fn.
"""
import numpy as np
import torch

__all__ = [
    'one_hot', 'mixup_target', 'rand_bbox', 'rand_bbox_minmax',
    'cutmix_bbox_and_lam', 'Mixup', 'FastCollateMixup',
]


def one_hot(x, num_classes, on_value=1., off_value=0.):
    x = x.long().view(-1, 1)
    dense = torch.full((x.shape[0], num_classes), off_value, device=x.device)
    return dense.scatter_(1, x, on_value)


def mixup_target(target, num_classes, lam=1., smoothing=0.0):
    """Soft targets: smoothed one-hot of target blended with its batch-flip."""
    off = smoothing / num_classes
    on = 1. - smoothing + off
    fwd = one_hot(target, num_classes, on_value=on, off_value=off)
    rev = one_hot(target.flip(0), num_classes, on_value=on, off_value=off)
    return fwd * lam + rev * (1. - lam)


def rand_bbox(img_shape, lam, margin=0., count=None):
    """CutMix box with area ratio (1-lam), centered at a uniform point."""
    ratio = np.sqrt(1 - lam)
    h, w = img_shape[-2:]
    cut_h, cut_w = int(h * ratio), int(w * ratio)
    pad_y, pad_x = int(margin * cut_h), int(margin * cut_w)
    cy = np.random.randint(pad_y, h - pad_y, size=count)
    cx = np.random.randint(pad_x, w - pad_x, size=count)
    return (
        np.clip(cy - cut_h // 2, 0, h),
        np.clip(cy + cut_h // 2, 0, h),
        np.clip(cx - cut_w // 2, 0, w),
        np.clip(cx + cut_w // 2, 0, w),
    )


def rand_bbox_minmax(img_shape, minmax, count=None):
    """CutMix box with side lengths drawn from a min/max ratio range."""
    assert len(minmax) == 2
    h, w = img_shape[-2:]
    cut_h = np.random.randint(int(h * minmax[0]), int(h * minmax[1]), size=count)
    cut_w = np.random.randint(int(w * minmax[0]), int(w * minmax[1]), size=count)
    yl = np.random.randint(0, h - cut_h, size=count)
    xl = np.random.randint(0, w - cut_w, size=count)
    return yl, yl + cut_h, xl, xl + cut_w


def cutmix_bbox_and_lam(img_shape, lam, ratio_minmax=None, correct_lam=True, count=None):
    """Draw the CutMix box; optionally recompute lambda from the clipped area."""
    if ratio_minmax is not None:
        box = rand_bbox_minmax(img_shape, ratio_minmax, count=count)
    else:
        box = rand_bbox(img_shape, lam, count=count)
    if correct_lam or ratio_minmax is not None:
        yl, yu, xl, xu = box
        lam = 1. - (yu - yl) * (xu - xl) / float(img_shape[-2] * img_shape[-1])
    return box, lam


class Mixup:
    """Mixup/CutMix over a collated batch, mode 'batch'|'pair'|'elem'.

    Applied on-device after the prefetcher; returns (mixed_x, soft_targets).
    """

    def __init__(
            self,
            mixup_alpha=1.,
            cutmix_alpha=0.,
            cutmix_minmax=None,
            prob=1.0,
            switch_prob=0.5,
            mode='batch',
            correct_lam=True,
            label_smoothing=0.1,
            num_classes=1000,
    ):
        self.mixup_alpha = mixup_alpha
        self.cutmix_alpha = cutmix_alpha
        self.cutmix_minmax = cutmix_minmax
        if cutmix_minmax is not None:
            assert len(cutmix_minmax) == 2
            self.cutmix_alpha = 1.0  # minmax boxes don't use the beta draw
        self.mix_prob = prob
        self.switch_prob = switch_prob
        self.mode = mode
        self.correct_lam = correct_lam
        self.label_smoothing = label_smoothing
        self.num_classes = num_classes
        self.mixup_enabled = True  # train loop may toggle off (e.g. final epochs)

    # -- parameter draws ----------------------------------------------------
    def _beta(self, alpha, size=None):
        return np.random.beta(alpha, alpha, size=size)

    def _draw_many(self, n):
        """Per-element draws: (lam[n] fp32, use_cutmix[n] bool)."""
        lam = np.ones(n, dtype=np.float32)
        use_cutmix = np.zeros(n, dtype=bool)
        if not self.mixup_enabled:
            return lam, use_cutmix
        has_mix, has_cut = self.mixup_alpha > 0., self.cutmix_alpha > 0.
        assert has_mix or has_cut, \
            'One of mixup_alpha > 0., cutmix_alpha > 0., cutmix_minmax not None should be true.'
        if has_mix and has_cut:
            use_cutmix = np.random.rand(n) < self.switch_prob
            drawn = np.where(
                use_cutmix,
                self._beta(self.cutmix_alpha, n),
                self._beta(self.mixup_alpha, n))
        elif has_mix:
            drawn = self._beta(self.mixup_alpha, n)
        else:
            use_cutmix = np.ones(n, dtype=bool)
            drawn = self._beta(self.cutmix_alpha, n)
        lam = np.where(np.random.rand(n) < self.mix_prob, drawn.astype(np.float32), lam)
        return lam, use_cutmix

    def _draw_one(self):
        """Whole-batch draw: (lam float, use_cutmix bool)."""
        if not (self.mixup_enabled and np.random.rand() < self.mix_prob):
            return 1., False
        has_mix, has_cut = self.mixup_alpha > 0., self.cutmix_alpha > 0.
        assert has_mix or has_cut, \
            'One of mixup_alpha > 0., cutmix_alpha > 0., cutmix_minmax not None should be true.'
        use_cutmix = has_cut and (not has_mix or np.random.rand() < self.switch_prob)
        lam = self._beta(self.cutmix_alpha if use_cutmix else self.mixup_alpha)
        return float(lam), use_cutmix

    def _cut_coords(self, shape, lam):
        return cutmix_bbox_and_lam(
            shape, lam, ratio_minmax=self.cutmix_minmax, correct_lam=self.correct_lam)

    # -- tensor mixing ------------------------------------------------------
    def _mix_elem(self, x):
        n = len(x)
        lam_all, use_cutmix = self._draw_many(n)
        source = x.clone()  # mixing sources must pre-date any in-place edits
        for i in range(n):
            j = n - i - 1
            lam = lam_all[i]
            if lam == 1.:
                continue
            if use_cutmix[i]:
                (yl, yh, xl, xh), lam_all[i] = self._cut_coords(x[i].shape, lam)
                x[i][:, yl:yh, xl:xh] = source[j][:, yl:yh, xl:xh]
            else:
                x[i] = x[i] * lam + source[j] * (1 - lam)
        return torch.tensor(lam_all, device=x.device, dtype=x.dtype).unsqueeze(1)

    def _mix_pair(self, x):
        n = len(x)
        lam_half, use_cutmix = self._draw_many(n // 2)
        source = x.clone()
        for i in range(n // 2):
            j = n - i - 1
            lam = lam_half[i]
            if lam == 1.:
                continue
            if use_cutmix[i]:
                (yl, yh, xl, xh), lam_half[i] = self._cut_coords(x[i].shape, lam)
                x[i][:, yl:yh, xl:xh] = source[j][:, yl:yh, xl:xh]
                x[j][:, yl:yh, xl:xh] = source[i][:, yl:yh, xl:xh]
            else:
                x[i] = x[i] * lam + source[j] * (1 - lam)
                x[j] = x[j] * lam + source[i] * (1 - lam)
        lam_full = np.concatenate((lam_half, lam_half[::-1]))
        return torch.tensor(lam_full, device=x.device, dtype=x.dtype).unsqueeze(1)

    def _mix_batch(self, x):
        lam, use_cutmix = self._draw_one()
        if lam == 1.:
            return 1.
        if use_cutmix:
            (yl, yh, xl, xh), lam = self._cut_coords(x.shape, lam)
            x[:, :, yl:yh, xl:xh] = x.flip(0)[:, :, yl:yh, xl:xh]
        else:
            flipped = x.flip(0).mul_(1. - lam)
            x.mul_(lam).add_(flipped)
        return lam

    def __call__(self, x, target):
        assert len(x) % 2 == 0, 'Batch size should be even when using this'
        mix = {'elem': self._mix_elem, 'pair': self._mix_pair}.get(self.mode, self._mix_batch)
        lam = mix(x)
        target = mixup_target(target, self.num_classes, lam, self.label_smoothing)
        return x, target


class FastCollateMixup(Mixup):
    """Mixup/CutMix executed on uint8 numpy frames inside the collate fn,
    before the batch tensor exists ('half' mode mixes only the first half)."""

    @staticmethod
    def _blend_u8(a, b, lam):
        out = a.astype(np.float32) * lam + b.astype(np.float32) * (1 - lam)
        np.rint(out, out=out)
        return out

    def _mix_elem_collate(self, output, batch, half=False):
        n_total = len(batch)
        n_out = n_total // 2 if half else n_total
        assert len(output) == n_out
        lam_all, use_cutmix = self._draw_many(n_out)
        for i in range(n_out):
            j = n_total - i - 1
            lam = lam_all[i]
            frame = batch[i][0]
            if lam != 1.:
                if use_cutmix[i]:
                    if not half:
                        frame = frame.copy()
                    (yl, yh, xl, xh), lam_all[i] = self._cut_coords(output.shape, lam)
                    frame[:, yl:yh, xl:xh] = batch[j][0][:, yl:yh, xl:xh]
                else:
                    frame = self._blend_u8(frame, batch[j][0], lam)
            output[i] += torch.from_numpy(frame.astype(np.uint8))
        if half:
            lam_all = np.concatenate((lam_all, np.ones(n_out)))
        return torch.tensor(lam_all).unsqueeze(1)

    def _mix_pair_collate(self, output, batch):
        n = len(batch)
        lam_half, use_cutmix = self._draw_many(n // 2)
        for i in range(n // 2):
            j = n - i - 1
            lam = lam_half[i]
            frame_i, frame_j = batch[i][0], batch[j][0]
            assert 0 <= lam <= 1.0
            if lam < 1.:
                if use_cutmix[i]:
                    (yl, yh, xl, xh), lam_half[i] = self._cut_coords(output.shape, lam)
                    patch = frame_i[:, yl:yh, xl:xh].copy()
                    frame_i[:, yl:yh, xl:xh] = frame_j[:, yl:yh, xl:xh]
                    frame_j[:, yl:yh, xl:xh] = patch
                else:
                    blended_i = self._blend_u8(frame_i, frame_j, lam)
                    frame_j = self._blend_u8(frame_j, frame_i, lam)
                    frame_i = blended_i
            output[i] += torch.from_numpy(frame_i.astype(np.uint8))
            output[j] += torch.from_numpy(frame_j.astype(np.uint8))
        lam_full = np.concatenate((lam_half, lam_half[::-1]))
        return torch.tensor(lam_full).unsqueeze(1)

    def _mix_batch_collate(self, output, batch):
        n = len(batch)
        lam, use_cutmix = self._draw_one()
        box = None
        if use_cutmix:
            box, lam = self._cut_coords(output.shape, lam)
        for i in range(n):
            j = n - i - 1
            frame = batch[i][0]
            if lam != 1.:
                if use_cutmix:
                    frame = frame.copy()  # sources iterate later; don't edit in place
                    yl, yh, xl, xh = box
                    frame[:, yl:yh, xl:xh] = batch[j][0][:, yl:yh, xl:xh]
                else:
                    frame = self._blend_u8(frame, batch[j][0], lam)
            output[i] += torch.from_numpy(frame.astype(np.uint8))
        return lam

    def __call__(self, batch, _=None):
        n = len(batch)
        assert n % 2 == 0, 'Batch size should be even when using this'
        half = 'half' in self.mode
        out_n = n // 2 if half else n
        output = torch.zeros((out_n, *batch[0][0].shape), dtype=torch.uint8)
        if self.mode in ('elem', 'half'):
            lam = self._mix_elem_collate(output, batch, half=half)
        elif self.mode == 'pair':
            lam = self._mix_pair_collate(output, batch)
        else:
            lam = self._mix_batch_collate(output, batch)
        target = torch.tensor([b[1] for b in batch], dtype=torch.int64)
        target = mixup_target(target, self.num_classes, lam, self.label_smoothing)
        return output, target[:out_n]
