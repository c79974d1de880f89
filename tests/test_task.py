"""Training-task layer tests (reference test surface: `tests/test_task.py`):
EMA lifecycle, checkpoint state round-trips (incl. task_state), no_sync
behavior outside distributed, and KD task loss composition."""
import copy

import pytest
import torch
import torch.nn as nn

import timm_amd
from timm_amd.task import (
    ClassificationTask, FeatureDistillationTask, LogitDistillationTask, TokenDistillationTask,
)


def _small_model(num_classes=10):
    torch.manual_seed(0)
    return timm_amd.create_model('resnet18', num_classes=num_classes)


def test_classification_task_forward_contract():
    task = ClassificationTask(_small_model(), criterion=nn.CrossEntropyLoss())
    out = task(torch.randn(2, 3, 64, 64), torch.randint(0, 10, (2,)))
    assert set(out) >= {'loss', 'output'}
    assert out['output'].shape == (2, 10)
    out['loss'].backward()
    assert next(task.model.parameters()).grad is not None


def test_task_ema_lifecycle():
    task = ClassificationTask(_small_model(), criterion=nn.CrossEntropyLoss())
    ema = task.setup_ema(decay=0.5, warmup=False)
    w0 = ema.module.conv1.weight.clone()
    with torch.no_grad():
        task.model.conv1.weight.add_(1.0)
    task.update_ema(step=5)
    w1 = ema.module.conv1.weight
    assert not torch.allclose(w0, w1)
    # step<=update_after_step+1 warm-starts with decay 0 (full copy)
    # decay 0.5: ema moves halfway toward the new weight
    expect = 0.5 * w0 + 0.5 * task.model.conv1.weight
    assert torch.allclose(w1, expect, atol=1e-5)


def test_task_checkpoint_roundtrip():
    task = ClassificationTask(_small_model(), criterion=nn.CrossEntropyLoss())
    task.setup_ema(decay=0.9, warmup=False)
    state = task.get_checkpoint_state()
    assert 'state_dict' in state and 'state_dict_ema' in state

    # mutate, then restore
    with torch.no_grad():
        for p in task.model.parameters():
            p.add_(1.0)
    task.load_checkpoint_state(state)
    reloaded = task.get_checkpoint_state()
    for k, v in state['state_dict'].items():
        assert torch.allclose(v.float(), reloaded['state_dict'][k].float()), k


def test_feature_distill_task_state_roundtrip():
    student = _small_model()
    teacher = timm_amd.create_model('resnet50', num_classes=10)
    task = FeatureDistillationTask(student, teacher, task_loss_weight=0.5, verbose=False)
    proj = task.model.projection
    assert proj is not None  # 512 -> 2048 projection
    state = task.get_checkpoint_state()
    assert 'task_state' in state and 'projection' in state['task_state']
    # plain student layout under state_dict
    assert 'conv1.weight' in state['state_dict']

    with torch.no_grad():
        proj.weight.add_(1.0)
    task.load_task_state(state['task_state'])
    assert torch.allclose(proj.weight, state['task_state']['projection']['weight'])


def test_no_sync_noop_outside_distributed():
    task = ClassificationTask(_small_model(), criterion=nn.CrossEntropyLoss())
    with task.no_sync():
        out = task(torch.randn(2, 3, 64, 64), torch.randint(0, 10, (2,)))
    out['loss'].backward()
    task.finish_gradient_sync()  # no distributed wrap: must be a no-op


def test_logit_distill_loss_composition():
    student = _small_model()
    teacher = copy.deepcopy(student)  # identical nets -> KL term == 0
    task = LogitDistillationTask(
        student, teacher, temperature=2.0, task_loss_weight=0.3, verbose=False)
    assert task.task_loss_weight == 0.3 and abs(task.distill_loss_weight - 0.7) < 1e-9
    student.eval()
    out = task(torch.randn(2, 3, 64, 64), torch.randint(0, 10, (2,)))
    assert out['distill_loss'].item() < 1e-5
    assert abs(out['loss'].item() - 0.3 * out['task_loss'].item()) < 1e-4


def test_token_distill_requires_distill_head():
    with pytest.raises(ValueError):
        TokenDistillationTask(_small_model(), _small_model(), verbose=False)
