from ._helpers import resume_task_checkpoint
from .classification import ClassificationTask
from .distillation import DistillationTeacher, LogitDistillationTask, FeatureDistillationTask
from .task import TrainingTask
