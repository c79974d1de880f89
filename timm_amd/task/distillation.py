"""Knowledge-distillation tasks (reference `timm/task/distillation.py`).

`DistillationTeacher` (:18) wraps a frozen teacher (re-normalizing inputs
between teacher/student data configs, :141); `LogitDistillationTask` (:201)
does temperature-scaled KL; `FeatureDistillationTask` (:471) matches
intermediate features through a learnable projection persisted in
task_state (:680).
"""
from typing import Dict, Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from .task import TrainingTask


class DistillationTeacher(nn.Module):
    """Frozen teacher wrapper w/ input re-normalization between data configs."""

    def __init__(
            self,
            model: nn.Module,
            student_mean: Optional[Tuple[float, ...]] = None,
            student_std: Optional[Tuple[float, ...]] = None,
            teacher_mean: Optional[Tuple[float, ...]] = None,
            teacher_std: Optional[Tuple[float, ...]] = None,
    ):
        super().__init__()
        self.model = model
        self.model.eval()
        for p in self.model.parameters():
            p.requires_grad_(False)

        # precompute renorm coefficients: x_t = (x_s * std_s + mean_s - mean_t) / std_t
        if student_mean is not None and teacher_mean is not None and (
                tuple(student_mean) != tuple(teacher_mean) or tuple(student_std) != tuple(teacher_std)):
            sm = torch.tensor(student_mean).view(1, -1, 1, 1)
            ss = torch.tensor(student_std).view(1, -1, 1, 1)
            tm = torch.tensor(teacher_mean).view(1, -1, 1, 1)
            ts = torch.tensor(teacher_std).view(1, -1, 1, 1)
            self.register_buffer('renorm_scale', ss / ts, persistent=False)
            self.register_buffer('renorm_bias', (sm - tm) / ts, persistent=False)
        else:
            self.renorm_scale = None
            self.renorm_bias = None

    @torch.no_grad()
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.renorm_scale is not None:
            x = x * self.renorm_scale.to(x.dtype) + self.renorm_bias.to(x.dtype)
        return self.model(x)

    def train(self, mode: bool = True):
        # teacher always stays in eval mode
        return super().train(False)


class LogitDistillationTask(TrainingTask):
    """KL(student || teacher) with temperature + CE mix (reference `:201`)."""

    def __init__(
            self,
            model: nn.Module,
            teacher: DistillationTeacher,
            criterion: Optional[nn.Module] = None,
            temperature: float = 4.0,
            alpha: float = 0.5,
    ):
        super().__init__()
        self.model = model
        self.teacher = teacher
        self.criterion = criterion if criterion is not None else nn.CrossEntropyLoss()
        self.temperature = temperature
        self.alpha = alpha

    def forward(self, input: torch.Tensor, target: torch.Tensor) -> Dict[str, torch.Tensor]:
        output = self.train_model(input)
        with torch.no_grad():
            teacher_output = self.teacher(input)
        T = self.temperature
        distill_loss = F.kl_div(
            F.log_softmax(output.float() / T, dim=-1),
            F.softmax(teacher_output.float() / T, dim=-1),
            reduction='batchmean',
        ) * (T * T)
        base_loss = self.criterion(output, target)
        loss = (1. - self.alpha) * base_loss + self.alpha * distill_loss
        return {'loss': loss, 'output': output, 'distill_loss': distill_loss, 'base_loss': base_loss}


class FeatureDistillationTask(TrainingTask):
    """Intermediate-feature matching distillation w/ learnable projection (reference `:471`)."""

    def __init__(
            self,
            model: nn.Module,
            teacher: DistillationTeacher,
            criterion: Optional[nn.Module] = None,
            alpha: float = 0.5,
            student_dim: Optional[int] = None,
            teacher_dim: Optional[int] = None,
            feature_loss: str = 'mse',
    ):
        super().__init__()
        self.model = model
        self.teacher = teacher
        self.criterion = criterion if criterion is not None else nn.CrossEntropyLoss()
        self.alpha = alpha
        self.feature_loss = feature_loss
        student_dim = student_dim or getattr(model, 'num_features', None)
        teacher_dim = teacher_dim or getattr(teacher.model, 'num_features', None)
        assert student_dim and teacher_dim
        self.projection = nn.Linear(student_dim, teacher_dim) if student_dim != teacher_dim else nn.Identity()

    def forward(self, input: torch.Tensor, target: torch.Tensor) -> Dict[str, torch.Tensor]:
        model = self.train_model
        inner = model.module if hasattr(model, 'module') else model
        feats = inner.forward_features(input)
        output = inner.forward_head(feats)
        with torch.no_grad():
            t_feats = self.teacher.model.forward_features(
                input if self.teacher.renorm_scale is None
                else input * self.teacher.renorm_scale.to(input.dtype) + self.teacher.renorm_bias.to(input.dtype))
        # pool token dims to compare global features
        s_pooled = feats.mean(dim=1) if feats.ndim == 3 else feats.mean(dim=(2, 3))
        t_pooled = t_feats.mean(dim=1) if t_feats.ndim == 3 else t_feats.mean(dim=(2, 3))
        s_proj = self.projection(s_pooled)
        if self.feature_loss == 'cosine':
            distill_loss = 1. - F.cosine_similarity(s_proj.float(), t_pooled.float(), dim=-1).mean()
        else:
            distill_loss = F.mse_loss(s_proj.float(), t_pooled.float())
        base_loss = self.criterion(output, target)
        loss = (1. - self.alpha) * base_loss + self.alpha * distill_loss
        return {'loss': loss, 'output': output, 'distill_loss': distill_loss, 'base_loss': base_loss}

    def get_task_state(self):
        return {'projection': self.projection.state_dict()}

    def load_task_state(self, state):
        if 'projection' in state:
            self.projection.load_state_dict(state['projection'])
