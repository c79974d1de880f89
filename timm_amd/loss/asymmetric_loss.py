"""Asymmetric focal losses (ASL, arxiv 2009.14119; reference
`timm/loss/asymmetric_loss.py`).

Positive and negative terms get separate focusing exponents, and negatives
are probability-shifted (clipped) so easy negatives drop out of the loss
entirely — the standard recipe for long-tailed multi-label training.
"""
import torch
import torch.nn as nn


class AsymmetricLossMultiLabel(nn.Module):
    def __init__(self, gamma_neg=4, gamma_pos=1, clip=0.05, eps=1e-8, disable_torch_grad_focal_loss=False):
        super().__init__()
        self.gamma_neg = gamma_neg
        self.gamma_pos = gamma_pos
        self.clip = clip
        self.eps = eps
        self.disable_torch_grad_focal_loss = disable_torch_grad_focal_loss

    def forward(self, x, y):
        """x: logits [B, C]; y: {0,1} multi-label targets [B, C]."""
        p_pos = torch.sigmoid(x)
        p_neg = 1 - p_pos
        if self.clip is not None and self.clip > 0:
            # probability shift: negatives with p < clip contribute nothing
            p_neg = (p_neg + self.clip).clamp(max=1)

        loss = y * torch.log(p_pos.clamp(min=self.eps)) \
            + (1 - y) * torch.log(p_neg.clamp(min=self.eps))

        if self.gamma_neg > 0 or self.gamma_pos > 0:
            if self.disable_torch_grad_focal_loss:
                torch.set_grad_enabled(False)
            pt = p_pos * y + p_neg * (1 - y)
            gamma = self.gamma_pos * y + self.gamma_neg * (1 - y)
            focal_w = torch.pow(1 - pt, gamma)
            if self.disable_torch_grad_focal_loss:
                torch.set_grad_enabled(True)
            loss = loss * focal_w

        return -loss.sum()


class AsymmetricLossSingleLabel(nn.Module):
    def __init__(self, gamma_pos=1, gamma_neg=4, eps: float = 0.1, reduction='mean'):
        super().__init__()
        self.gamma_pos = gamma_pos
        self.gamma_neg = gamma_neg
        self.eps = eps
        self.reduction = reduction
        self.logsoftmax = nn.LogSoftmax(dim=-1)
        self.targets_classes = []  # kept as attribute for reference API parity

    def forward(self, inputs, target):
        """inputs: logits [B, C]; target: int labels [B]."""
        num_classes = inputs.size(-1)
        log_preds = self.logsoftmax(inputs)
        onehot = torch.zeros_like(inputs).scatter_(1, target.long().unsqueeze(1), 1)
        self.targets_classes = onehot

        inv = 1 - onehot
        p = torch.exp(log_preds)
        pt = p * onehot + (1 - p) * inv
        gamma = self.gamma_pos * onehot + self.gamma_neg * inv
        log_preds = log_preds * torch.pow(1 - pt, gamma)

        if self.eps > 0:  # label smoothing
            onehot = onehot.mul(1 - self.eps).add(self.eps / num_classes)
            self.targets_classes = onehot

        loss = -onehot.mul(log_preds).sum(dim=-1)
        if self.reduction == 'mean':
            loss = loss.mean()
        return loss
