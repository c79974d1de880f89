"""Classification training task (reference `timm/task/classification.py:13`)."""
from typing import Dict, Optional

import torch
import torch.nn as nn

from .task import TrainingTask


class ClassificationTask(TrainingTask):
    """Standard supervised classification: model(input) -> criterion(output, target)."""

    def __init__(
            self,
            model: nn.Module,
            criterion: Optional[nn.Module] = None,
    ):
        super().__init__()
        self.model = model
        self.criterion = criterion if criterion is not None else nn.CrossEntropyLoss()

    def forward(self, input: torch.Tensor, target: torch.Tensor) -> Dict[str, torch.Tensor]:
        output = self.train_model(input)
        loss = self.criterion(output, target)
        return {'loss': loss, 'output': output}
