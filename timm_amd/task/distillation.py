"""Knowledge-distillation tasks (reference `timm/task/distillation.py`).

`DistillationTeacher` (:18) wraps a frozen teacher and re-normalizes inputs
between the student's and teacher's data configs. `LogitDistillationTask`
(:201) does temperature-scaled KL on the output logits;
`FeatureDistillationTask` (:471) matches pre-logits features through a
learnable projection that trains (and all-reduces) together with the student
and persists in `task_state`.
"""
import logging
from typing import Dict, Optional, Tuple, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..utils.model import unwrap_model
from .task import TrainingTask

_logger = logging.getLogger(__name__)

__all__ = ['DistillationTeacher', 'LogitDistillationTask', 'FeatureDistillationTask']


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

    def renorm(self, x: torch.Tensor) -> torch.Tensor:
        if self.renorm_scale is None:
            return x
        return x * self.renorm_scale.to(x.dtype) + self.renorm_bias.to(x.dtype)

    @torch.no_grad()
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.model(self.renorm(x))

    @torch.no_grad()
    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        return self.model.forward_features(self.renorm(x))

    def train(self, mode: bool = True):
        # teacher always stays in eval mode
        return super().train(False)


def _norm_cfg(model: nn.Module):
    cfg = getattr(unwrap_model(model), 'pretrained_cfg', None) or {}
    return cfg.get('mean', (0.485, 0.456, 0.406)), cfg.get('std', (0.229, 0.224, 0.225))


def _resolve_teacher(
        teacher_model: Union[str, nn.Module, DistillationTeacher],
        student_model: nn.Module,
        pretrained_path: Optional[str] = None,
        device: Optional[torch.device] = None,
        dtype: Optional[torch.dtype] = None,
) -> DistillationTeacher:
    """Accept a model name / module / wrapped teacher (reference `:141`)."""
    if isinstance(teacher_model, DistillationTeacher):
        return teacher_model
    if isinstance(teacher_model, str):
        from ..models import create_model
        student = unwrap_model(student_model)
        _logger.info(f"Creating distillation teacher model: '{teacher_model}'")
        kwargs = {'pretrained': True}
        if pretrained_path:
            kwargs['pretrained_cfg_overlay'] = dict(file=pretrained_path, num_classes=student.num_classes)
        module = create_model(
            teacher_model,
            num_classes=student.num_classes,
            in_chans=getattr(student, 'in_chans', 3),
            **kwargs,
        )
        if device is not None or dtype is not None:
            module = module.to(device=device, dtype=dtype)
    elif isinstance(teacher_model, nn.Module):
        module = teacher_model
    else:
        raise TypeError(
            f'teacher_model must be a name, nn.Module or DistillationTeacher, '
            f'got {type(teacher_model).__name__}')
    s_mean, s_std = _norm_cfg(student_model)
    t_mean, t_std = _norm_cfg(module)
    return DistillationTeacher(
        module, student_mean=s_mean, student_std=s_std, teacher_mean=t_mean, teacher_std=t_std)


def _resolve_loss_weights(
        distill_loss_weight: Optional[float],
        task_loss_weight: Optional[float],
) -> Tuple[float, float]:
    """(task_w, distill_w): both given -> independent; task only ->
    complementary; neither -> 1/1 (reference `:252-278`)."""
    if distill_loss_weight is not None:
        return (task_loss_weight if task_loss_weight is not None else 1.0), distill_loss_weight
    if task_loss_weight is not None:
        return task_loss_weight, 1.0 - task_loss_weight
    return 1.0, 1.0


class LogitDistillationTask(TrainingTask):
    """KL(student || teacher) on output logits with temperature (reference `:201`)."""

    def __init__(
            self,
            student_model: nn.Module,
            teacher_model: Union[str, nn.Module, DistillationTeacher],
            criterion: Optional[nn.Module] = None,
            teacher_pretrained_path: Optional[str] = None,
            loss_type: str = 'kl',
            distill_loss_weight: Optional[float] = None,
            task_loss_weight: Optional[float] = None,
            temperature: float = 1.0,
            device: Optional[torch.device] = None,
            dtype: Optional[torch.dtype] = None,
            verbose: bool = True,
    ):
        super().__init__()
        if loss_type != 'kl':
            raise ValueError(f"Unsupported loss_type '{loss_type}'. Currently only 'kl' is supported.")
        self.model = student_model
        self.teacher = _resolve_teacher(teacher_model, student_model, teacher_pretrained_path, device, dtype)
        self.criterion = criterion if criterion is not None else nn.CrossEntropyLoss()
        self.loss_type = loss_type
        self.temperature = temperature
        self.task_loss_weight, self.distill_loss_weight = _resolve_loss_weights(
            distill_loss_weight, task_loss_weight)
        if verbose:
            _logger.info(
                f'LogitDistillationTask: T={temperature} '
                f'task_w={self.task_loss_weight} distill_w={self.distill_loss_weight}')

    def forward(self, input: torch.Tensor, target: torch.Tensor) -> Dict[str, torch.Tensor]:
        output = self.train_model(input)
        with torch.no_grad():
            teacher_output = self.teacher(input)
        T = self.temperature
        distill_loss = F.kl_div(
            F.log_softmax(output.float() / T, dim=-1),
            F.log_softmax(teacher_output.float() / T, dim=-1),
            reduction='batchmean',
            log_target=True,
        ) * (T * T)
        task_loss = self.criterion(output, target)
        loss = self.task_loss_weight * task_loss + self.distill_loss_weight * distill_loss
        return {
            'loss': loss,
            'output': output,
            'task_loss': task_loss,
            'distill_loss': distill_loss,
            # kept for older callers
            'base_loss': task_loss,
        }


class _StudentWithProjection(nn.Module):
    """Student + feature projection as ONE trainable module so the bucketed
    all-reduce covers the projection grads too (reference `:415`)."""

    def __init__(self, student: nn.Module, projection: Optional[nn.Module]):
        super().__init__()
        self.student = student
        self.projection = projection

    def forward(self, x: torch.Tensor):
        feats = self.student.forward_features(x)
        output = self.student.forward_head(feats)
        pooled = feats.mean(dim=1) if feats.ndim == 3 else feats.mean(dim=(2, 3))
        if self.projection is not None:
            pooled = self.projection(pooled)
        return output, pooled


class FeatureDistillationTask(TrainingTask):
    """Pre-logits feature matching through a learnable projection (reference `:471`)."""

    def __init__(
            self,
            student_model: nn.Module,
            teacher_model: Union[str, nn.Module, DistillationTeacher],
            criterion: Optional[nn.Module] = None,
            teacher_pretrained_path: Optional[str] = None,
            distill_loss_weight: Optional[float] = None,
            task_loss_weight: Optional[float] = None,
            student_feature_dim: Optional[int] = None,
            teacher_feature_dim: Optional[int] = None,
            feature_loss: str = 'mse',
            device: Optional[torch.device] = None,
            dtype: Optional[torch.dtype] = None,
            verbose: bool = True,
    ):
        super().__init__()
        self.teacher = _resolve_teacher(teacher_model, student_model, teacher_pretrained_path, device, dtype)
        self.criterion = criterion if criterion is not None else nn.CrossEntropyLoss()
        self.feature_loss = feature_loss
        self.task_loss_weight, self.distill_loss_weight = _resolve_loss_weights(
            distill_loss_weight, task_loss_weight)

        if student_feature_dim is None:
            student_feature_dim = self._detect_feature_dim(student_model)
        if teacher_feature_dim is None:
            teacher_feature_dim = self._detect_feature_dim(self.teacher.model)
        projection = None
        if student_feature_dim != teacher_feature_dim:
            if verbose:
                _logger.info(f'Creating projection layer: {student_feature_dim} -> {teacher_feature_dim}')
            projection = nn.Linear(student_feature_dim, teacher_feature_dim)
            if device is not None or dtype is not None:
                projection = projection.to(device=device, dtype=dtype)
        self.model = _StudentWithProjection(student_model, projection)
        if verbose:
            _logger.info(
                f'FeatureDistillationTask: student_dim={student_feature_dim} '
                f'teacher_dim={teacher_feature_dim} '
                f'task_w={self.task_loss_weight} distill_w={self.distill_loss_weight}')

    @staticmethod
    def _detect_feature_dim(model: nn.Module) -> int:
        model = unwrap_model(model)
        if hasattr(model, 'head_hidden_size'):
            return model.head_hidden_size
        if hasattr(model, 'num_features'):
            return model.num_features
        raise ValueError(
            'Cannot auto-detect feature dimension: model needs head_hidden_size or '
            'num_features, or pass student_feature_dim/teacher_feature_dim.')

    @property
    def eval_model(self) -> nn.Module:
        """The plain student (no projection wrapper) for validation."""
        return unwrap_model(self.model).student

    def forward(self, input: torch.Tensor, target: torch.Tensor) -> Dict[str, torch.Tensor]:
        output, s_proj = self.train_model(input)
        with torch.no_grad():
            t_feats = self.teacher.forward_features(input)
        t_pooled = t_feats.mean(dim=1) if t_feats.ndim == 3 else t_feats.mean(dim=(2, 3))
        if self.feature_loss == 'cosine':
            distill_loss = 1. - F.cosine_similarity(s_proj.float(), t_pooled.float(), dim=-1).mean()
        else:
            distill_loss = F.mse_loss(s_proj.float(), t_pooled.float())
        task_loss = self.criterion(output, target)
        loss = self.task_loss_weight * task_loss + self.distill_loss_weight * distill_loss
        return {
            'loss': loss,
            'output': output,
            'task_loss': task_loss,
            'distill_loss': distill_loss,
            'base_loss': task_loss,
        }

    def get_checkpoint_state(self):
        # persist the plain student under 'state_dict' (same layout as a
        # non-KD run); the projection rides in 'task_state'
        state = {'state_dict': unwrap_model(self.eval_model).state_dict()}
        if self.model_ema is not None:
            state['state_dict_ema'] = unwrap_model(self.model_ema.module).state_dict()
        task_state = self.get_task_state()
        if task_state:
            state['task_state'] = task_state
        return state

    def get_task_state(self):
        projection = unwrap_model(self.model).projection
        if projection is None:
            return {}
        return {'projection': projection.state_dict()}

    def load_task_state(self, state):
        projection = unwrap_model(self.model).projection
        if 'projection' in state and projection is not None:
            projection.load_state_dict(state['projection'])
