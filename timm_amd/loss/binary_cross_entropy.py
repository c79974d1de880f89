"""Binary cross-entropy over class logits with dense (Mixup/CutMix) targets.

Behavioral parity: /root/reference/timm/loss/binary_cross_entropy.py
(smoothing-as-densify, target thresholding, sum-classes reduction,
weight/pos_weight buffers).
"""
from typing import Optional, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = ['BinaryCrossEntropy']


class BinaryCrossEntropy(nn.Module):
    """BCE-with-logits for classification.

    Sparse integer targets are densified to one-hot with label smoothing
    (smoothing assumed already applied upstream for dense targets).  With
    ``target_threshold`` dense targets are binarized; with ``sum_classes`` the
    per-class losses are summed before the batch mean.
    """

    def __init__(
            self,
            smoothing=0.1,
            target_threshold: Optional[float] = None,
            weight: Optional[torch.Tensor] = None,
            reduction: str = 'mean',
            sum_classes: bool = False,
            pos_weight: Optional[Union[torch.Tensor, float]] = None,
    ):
        super().__init__()
        assert 0. <= smoothing < 1.0
        self.smoothing = smoothing
        self.target_threshold = target_threshold
        self.sum_classes = sum_classes
        self.reduction = 'none' if sum_classes else reduction
        self.register_buffer('weight', weight)
        if pos_weight is not None and not isinstance(pos_weight, torch.Tensor):
            pos_weight = torch.tensor(pos_weight)
        self.register_buffer('pos_weight', pos_weight)

    def _densify(self, target: torch.Tensor, num_classes: int, like: torch.Tensor):
        low = self.smoothing / num_classes
        high = 1. - self.smoothing + low
        dense = torch.full(
            (target.shape[0], num_classes), low, device=like.device, dtype=like.dtype)
        return dense.scatter_(1, target.long().view(-1, 1), high)

    def forward(self, x: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        assert x.shape[0] == target.shape[0]
        if target.shape != x.shape:
            target = self._densify(target, x.shape[-1], x)
        if self.target_threshold is not None:
            target = target.gt(self.target_threshold).to(dtype=target.dtype)
        loss = F.binary_cross_entropy_with_logits(
            x, target, self.weight,
            pos_weight=self.pos_weight,
            reduction=self.reduction,
        )
        if self.sum_classes:
            loss = loss.sum(-1).mean()
        return loss
