"""AMP loss-scaler wrapper (reference `timm/utils/cuda.py:46`).

On MI355X the benchmark dtype is bf16 (no scaler needed), but fp16 AMP
training keeps the reference's NativeScaler semantics: scale -> backward ->
unscale -> clip -> step.
"""
from typing import Optional

import torch

from .clip_grad import dispatch_clip_grad


class NativeScaler:
    state_dict_key = "amp_scaler"

    def __init__(self, device='cuda'):
        self._scaler = torch.amp.GradScaler(device=device)

    def __call__(
            self,
            loss,
            optimizer,
            clip_grad=None,
            clip_mode='norm',
            parameters=None,
            create_graph=False,
            need_update=True,
            pre_step_fn=None,
    ):
        self._scaler.scale(loss).backward(create_graph=create_graph)
        if need_update:
            if pre_step_fn is not None:
                # e.g. task.finish_gradient_sync: wait on in-flight bucket
                # all-reduces + apply 1/world averaging BEFORE unscale/step
                pre_step_fn()
            if clip_grad is not None:
                assert parameters is not None
                self._scaler.unscale_(optimizer)  # unscale the gradients of optimizer's assigned params in-place
                dispatch_clip_grad(parameters, clip_grad, mode=clip_mode)
            self._scaler.step(optimizer)
            self._scaler.update()

    def state_dict(self):
        return self._scaler.state_dict()

    def load_state_dict(self, state_dict):
        self._scaler.load_state_dict(state_dict)
