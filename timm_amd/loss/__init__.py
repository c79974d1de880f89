"""Training losses.

All CE-family losses route through the fused HIP cross-entropy kernel on
GPU tensors (see `timm_amd/ops/loss.py`).
"""
from .cross_entropy import LabelSmoothingCrossEntropy, SoftTargetCrossEntropy
from .binary_cross_entropy import BinaryCrossEntropy
from .asymmetric_loss import AsymmetricLossMultiLabel, AsymmetricLossSingleLabel
from .jsd import JsdCrossEntropy

__all__ = [
    'LabelSmoothingCrossEntropy', 'SoftTargetCrossEntropy', 'BinaryCrossEntropy',
    'AsymmetricLossMultiLabel', 'AsymmetricLossSingleLabel', 'JsdCrossEntropy',
]
