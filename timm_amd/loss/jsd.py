"""AugMix consistency loss: clean-split CE + Jensen-Shannon divergence across
augmentation splits.  Behavioral parity: /root/reference/timm/loss/jsd.py.
"""
import torch
import torch.nn as nn
import torch.nn.functional as F

from .cross_entropy import LabelSmoothingCrossEntropy

__all__ = ['JsdCrossEntropy']


class JsdCrossEntropy(nn.Module):
    """CE on the clean split + alpha * JSD(clean, aug1, ..., augN-1).

    Expects the batch stacked as num_splits equal chunks (clean first), the
    AugMixDataset layout.
    """

    def __init__(self, num_splits=3, alpha=12, smoothing=0.1):
        super().__init__()
        self.num_splits = num_splits
        self.alpha = alpha
        self.ce = (
            LabelSmoothingCrossEntropy(smoothing)
            if smoothing else nn.CrossEntropyLoss()
        )

    def forward(self, output, target):
        n = output.shape[0] // self.num_splits
        assert n * self.num_splits == output.shape[0]
        chunks = output.split(n)

        # supervised term sees only the unaugmented first chunk
        loss = self.ce(chunks[0], target[:n])

        # JSD term: mean KL of each split distribution to the mixture
        dists = [F.softmax(c, dim=1) for c in chunks]
        log_mix = torch.stack(dists).mean(dim=0).clamp(1e-7, 1).log()
        jsd = sum(F.kl_div(log_mix, d, reduction='batchmean') for d in dists)
        return loss + self.alpha * jsd / len(dists)
