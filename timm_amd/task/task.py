"""TrainingTask base (reference `timm/task/task.py:17`).

Owns the trainable module + criterion.  `prepare_distributed()` wraps with
our BucketedDataParallel (RCCL/xGMI reducer) instead of torch DDP;
`no_sync()` proxies the reducer's grad-accumulation context.
Checkpoint state keeps the reference key layout (`state_dict`,
`state_dict_ema`, `task_state`).
"""
import contextlib
from typing import Any, Dict, Optional

import torch
import torch.nn as nn

from ..utils.model import unwrap_model
from ..utils.model_ema import ModelEmaV3


class TrainingTask(nn.Module):
    """Base class for training tasks: forward(input, target) -> {'loss', 'output'}."""

    def __init__(self):
        super().__init__()
        self.model: Optional[nn.Module] = None
        self.model_ema: Optional[ModelEmaV3] = None
        self._distributed_model = None
        self._compiled = False

    # -- core API -----------------------------------------------------------

    def forward(self, input: torch.Tensor, target: torch.Tensor) -> Dict[str, torch.Tensor]:
        raise NotImplementedError

    @property
    def train_model(self) -> nn.Module:
        """The module actually called in the train step (DDP-wrapped if distributed)."""
        return self._distributed_model if self._distributed_model is not None else self.model

    def prepare_distributed(self, device_ids=None, bucket_cap_mb: float = 50., **kwargs):
        """Wrap the trainable module for data-parallel training over RCCL."""
        from ..parallel import BucketedDataParallel
        self._distributed_model = BucketedDataParallel(self.model, bucket_cap_mb=bucket_cap_mb, **kwargs)
        return self._distributed_model

    def compile(self, backend: str = 'inductor', mode: Optional[str] = None):
        """torch.compile the model (called BEFORE prepare_distributed, reference `train.py:1015`)."""
        assert self._distributed_model is None, 'compile() must be called before prepare_distributed()'
        self.model = torch.compile(self.model, backend=backend, mode=mode)
        self._compiled = True
        return self.model

    # -- EMA ----------------------------------------------------------------

    def setup_ema(
            self,
            decay: float = 0.9998,
            warmup: bool = True,
            device: Optional[torch.device] = None,
            force_cpu: bool = False,
    ):
        """Create the EMA copy of the model (reference `task.py:108`)."""
        self.model_ema = ModelEmaV3(
            unwrap_model(self.model),
            decay=decay,
            use_warmup=warmup,
            device='cpu' if force_cpu else device,
        )
        return self.model_ema

    def update_ema(self, step: Optional[int] = None):
        if self.model_ema is not None:
            self.model_ema.update(unwrap_model(self.model), step=step)

    def compile_ema(self, backend: str = 'inductor'):
        if self.model_ema is not None:
            self.model_ema.module.compile(backend=backend)

    # -- grad sync ----------------------------------------------------------

    @contextlib.contextmanager
    def no_sync(self):
        """Grad-accumulation context (no all-reduce), reference `task.py:231`."""
        if self._distributed_model is not None:
            with self._distributed_model.no_sync():
                yield
        else:
            yield

    def finish_gradient_sync(self):
        if self._distributed_model is not None:
            self._distributed_model.finish_gradient_sync()

    # -- checkpoint ---------------------------------------------------------

    def get_checkpoint_state(self) -> Dict[str, Any]:
        """Emit checkpoint entries in reference layout (reference `task.py:191-210`)."""
        state = {'state_dict': unwrap_model(self.model).state_dict()}
        if self.model_ema is not None:
            state['state_dict_ema'] = unwrap_model(self.model_ema.module).state_dict()
        task_state = self.get_task_state()
        if task_state:
            state['task_state'] = task_state
        return state

    def load_checkpoint_state(self, state: Dict[str, Any]):
        from ..models import clean_state_dict
        if 'state_dict' in state:
            unwrap_model(self.model).load_state_dict(clean_state_dict(state['state_dict']))
        if 'state_dict_ema' in state and self.model_ema is not None:
            unwrap_model(self.model_ema.module).load_state_dict(clean_state_dict(state['state_dict_ema']))
        if 'task_state' in state:
            self.load_task_state(state['task_state'])

    def get_task_state(self) -> Dict[str, Any]:
        """Override to persist task-specific state (e.g. distill projections)."""
        return {}

    def load_task_state(self, state: Dict[str, Any]):
        pass
