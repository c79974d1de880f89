"""Task checkpoint resume helpers (reference `timm/task/_helpers.py:23-104`)."""
import logging
import os
from typing import Any, Optional

import torch

from ..models._helpers import _torch_load, clean_state_dict

_logger = logging.getLogger(__name__)


def resume_task_checkpoint(
        task,
        checkpoint_path: str,
        optimizer: Optional[torch.optim.Optimizer] = None,
        loss_scaler: Optional[Any] = None,
        log_info: bool = True,
):
    """Resume full train state into a TrainingTask: model (+EMA), optimizer, scaler, epoch."""
    resume_epoch = None
    if not os.path.isfile(checkpoint_path):
        _logger.error(f"No checkpoint found at '{checkpoint_path}'")
        raise FileNotFoundError(checkpoint_path)

    checkpoint = _torch_load(checkpoint_path, map_location='cpu', weights_only=False)
    if isinstance(checkpoint, dict) and 'state_dict' in checkpoint:
        if log_info:
            _logger.info('Restoring model state from checkpoint...')
        task.load_checkpoint_state(checkpoint)

        if optimizer is not None and 'optimizer' in checkpoint:
            if log_info:
                _logger.info('Restoring optimizer state from checkpoint...')
            optimizer.load_state_dict(checkpoint['optimizer'])

        if loss_scaler is not None and getattr(loss_scaler, 'state_dict_key', None) in checkpoint:
            if log_info:
                _logger.info('Restoring AMP loss scaler state from checkpoint...')
            loss_scaler.load_state_dict(checkpoint[loss_scaler.state_dict_key])

        if 'epoch' in checkpoint:
            resume_epoch = checkpoint['epoch']
            if 'version' in checkpoint and checkpoint['version'] > 1:
                resume_epoch += 1  # start at the next epoch, old checkpoints incremented before save
        if log_info:
            _logger.info(f"Loaded checkpoint '{checkpoint_path}' (epoch {checkpoint.get('epoch', '?')})")
    else:
        from ..utils.model import unwrap_model
        unwrap_model(task.model).load_state_dict(clean_state_dict(checkpoint))
        if log_info:
            _logger.info(f"Loaded checkpoint '{checkpoint_path}'")
    return resume_epoch
