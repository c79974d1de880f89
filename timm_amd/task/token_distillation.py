"""Token-based distillation for models with a dedicated distillation head
(reference `timm/task/token_distillation.py:17-368`).

DeiT-style students expose `set_distilled_training(True)` and then return a
`(class_logits, distill_logits)` tuple: the class head trains against the
labels while the distill head matches the (frozen) teacher — either its soft
temperature-scaled distribution or its hard argmax.
"""
import logging
from typing import Dict, Optional, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..utils.model import unwrap_model
from .task import TrainingTask

_logger = logging.getLogger(__name__)

__all__ = ['TokenDistillationTeacher', 'TokenDistillationTask']


class TokenDistillationTeacher(nn.Module):
    """Frozen teacher for token distillation; holds the teacher's input
    normalization so student-normalized batches can be re-normalized."""

    def __init__(
            self,
            model_name_or_module: Union[str, nn.Module],
            num_classes: Optional[int] = None,
            in_chans: int = 3,
            pretrained_path: Optional[str] = None,
            device: Optional[torch.device] = None,
            dtype: Optional[torch.dtype] = None,
    ):
        super().__init__()
        if isinstance(model_name_or_module, str):
            from ..models import create_model
            _logger.info(f"Creating token distillation teacher model: '{model_name_or_module}'")
            kwargs = {'pretrained': True}
            if pretrained_path:
                kwargs['pretrained_cfg_overlay'] = dict(file=pretrained_path, num_classes=num_classes)
            model = create_model(
                model_name_or_module, num_classes=num_classes, in_chans=in_chans, **kwargs)
            if device is not None or dtype is not None:
                model = model.to(device=device, dtype=dtype)
        elif isinstance(model_name_or_module, nn.Module):
            model = model_name_or_module
        else:
            raise TypeError(
                f'model_name_or_module must be a string or nn.Module, got {type(model_name_or_module).__name__}')

        model.eval()
        self.model = model

        inner = unwrap_model(model)
        cfg = getattr(inner, 'pretrained_cfg', None) or {}
        mean = cfg.get('mean', (0.485, 0.456, 0.406))
        std = cfg.get('std', (0.229, 0.224, 0.225))
        self.register_buffer(
            'mean_kd', torch.tensor(mean, device=device, dtype=dtype).view(1, -1, 1, 1), persistent=False)
        self.register_buffer(
            'std_kd', torch.tensor(std, device=device, dtype=dtype).view(1, -1, 1, 1), persistent=False)

    def compile(self, backend: str = 'inductor', mode: Optional[str] = None, **kwargs):
        self.model = torch.compile(self.model, backend=backend, mode=mode, **kwargs)
        return self

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        return self.model(input)

    def normalize_input(
            self,
            input: torch.Tensor,
            student_mean: Optional[torch.Tensor] = None,
            student_std: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        """Map a student-normalized batch onto the teacher's normalization."""
        if student_mean is None or student_std is None:
            return input
        if torch.equal(student_mean, self.mean_kd) and torch.equal(student_std, self.std_kd):
            return input
        return (input * student_std + student_mean - self.mean_kd) / self.std_kd

    def train(self, mode: bool = True):
        return super().train(False)  # teacher never leaves eval


class TokenDistillationTask(TrainingTask):
    """Class head vs labels + distill head vs teacher (soft KL or hard CE).

    Weighting (reference `:252-278`): both weights given -> independent;
    only task weight -> complementary (distill = 1 - task); neither -> 1/1.
    """

    def __init__(
            self,
            student_model: nn.Module,
            teacher_model: Union[str, nn.Module, TokenDistillationTeacher],
            criterion: Optional[nn.Module] = None,
            teacher_pretrained_path: Optional[str] = None,
            distill_type: str = 'soft',
            distill_loss_weight: Optional[float] = None,
            task_loss_weight: Optional[float] = None,
            temperature: float = 1.0,
            device: Optional[torch.device] = None,
            dtype: Optional[torch.dtype] = None,
            verbose: bool = True,
    ):
        super().__init__()
        if distill_type not in ('soft', 'hard'):
            raise ValueError(f"Unsupported distill_type '{distill_type}'. Must be 'soft' or 'hard'.")

        student = unwrap_model(student_model)
        if not hasattr(student, 'set_distilled_training'):
            raise ValueError(
                f"Model {student.__class__.__name__} has no 'set_distilled_training'; "
                'TokenDistillationTask needs a distillation-token model (e.g. deit_*_distilled).')
        student.set_distilled_training(True)

        if isinstance(teacher_model, TokenDistillationTeacher):
            teacher = teacher_model
        elif isinstance(teacher_model, (str, nn.Module)):
            teacher = TokenDistillationTeacher(
                teacher_model,
                num_classes=student.num_classes,
                in_chans=getattr(student, 'in_chans', 3),
                pretrained_path=teacher_pretrained_path,
                device=device,
                dtype=dtype,
            )
        else:
            raise TypeError(
                f'teacher_model must be a name, nn.Module or TokenDistillationTeacher, '
                f'got {type(teacher_model).__name__}')

        self.model = student_model
        self.teacher = teacher
        self.criterion = criterion if criterion is not None else nn.CrossEntropyLoss()
        self.distill_type = distill_type
        self.temperature = temperature

        cfg = getattr(student, 'pretrained_cfg', None) or {}
        self.register_buffer(
            'student_mean',
            torch.tensor(cfg.get('mean', (0.485, 0.456, 0.406)), device=device, dtype=dtype).view(1, -1, 1, 1),
            persistent=False)
        self.register_buffer(
            'student_std',
            torch.tensor(cfg.get('std', (0.229, 0.224, 0.225)), device=device, dtype=dtype).view(1, -1, 1, 1),
            persistent=False)

        if distill_loss_weight is not None:
            self.distill_loss_weight = distill_loss_weight
            self.task_loss_weight = task_loss_weight if task_loss_weight is not None else 1.0
        elif task_loss_weight is not None:
            self.task_loss_weight = task_loss_weight
            self.distill_loss_weight = 1.0 - task_loss_weight
        else:
            self.distill_loss_weight = 1.0
            self.task_loss_weight = 1.0
        if verbose:
            _logger.info(
                f'TokenDistillationTask: type={distill_type} T={temperature} '
                f'task_w={self.task_loss_weight} distill_w={self.distill_loss_weight}')

    def forward(self, input: torch.Tensor, target: torch.Tensor) -> Dict[str, torch.Tensor]:
        main_logits, dist_logits = self.train_model(input)
        task_loss = self.criterion(main_logits, target)

        with torch.no_grad():
            input_kd = self.teacher.normalize_input(input, self.student_mean, self.student_std)
            teacher_logits = self.teacher(input_kd.detach())

        if self.distill_type == 'soft':
            T = self.temperature
            distill_loss = F.kl_div(
                F.log_softmax(dist_logits / T, dim=-1),
                F.log_softmax(teacher_logits / T, dim=-1),
                reduction='batchmean',
                log_target=True,
            ) * (T * T)
        else:
            distill_loss = F.cross_entropy(dist_logits, teacher_logits.argmax(dim=-1))

        loss = self.task_loss_weight * task_loss + self.distill_loss_weight * distill_loss
        return {
            'loss': loss,
            'output': main_logits,
            'task_loss': task_loss,
            'distill_loss': distill_loss,
        }
