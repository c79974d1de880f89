"""Label-smoothing + soft-target CE (reference `timm/loss/cross_entropy.py`).

On ROCm both modules route through the fused HIP kernel
(`ops/loss.py` / `csrc/ce_loss.hip`): one pass over the logits per direction,
softmax recomputed from LSE in backward. CPU keeps the composable form.
"""
import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops


class LabelSmoothingCrossEntropy(nn.Module):
    """NLL loss with label smoothing."""

    def __init__(self, smoothing=0.1):
        super().__init__()
        assert smoothing < 1.0
        self.smoothing = smoothing
        self.confidence = 1. - smoothing

    def forward(self, x: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        if x.is_cuda and x.dim() == 2 and target.dim() == 1:
            return ops.fused_cross_entropy(x, target, smoothing=self.smoothing)
        logprobs = F.log_softmax(x, dim=-1)
        nll_loss = -logprobs.gather(dim=-1, index=target.unsqueeze(1))
        nll_loss = nll_loss.squeeze(1)
        smooth_loss = -logprobs.mean(dim=-1)
        loss = self.confidence * nll_loss + self.smoothing * smooth_loss
        return loss.mean()


class SoftTargetCrossEntropy(nn.Module):

    def __init__(self):
        super().__init__()

    def forward(self, x: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        if x.is_cuda and x.dim() == 2 and target.shape == x.shape:
            return ops.fused_cross_entropy(x, target.to(x.dtype), smoothing=0.)
        loss = torch.sum(-target * F.log_softmax(x, dim=-1), dim=-1)
        return loss.mean()
