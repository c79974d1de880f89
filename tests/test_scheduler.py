"""Value-curve tests for the LR scheduler core (reference test style:
/root/reference/tests/test_scheduler.py — warmup monotonicity, cosine
endpoints, cycles, k-decay, noise bounds, state-dict round-trip, lr_scale)."""
import math

import pytest
import torch

from timm_amd.scheduler import (
    CosineLRScheduler,
    MultiStepLRScheduler,
    PlateauLRScheduler,
    PolyLRScheduler,
    StepLRScheduler,
    TanhLRScheduler,
    create_scheduler_v2,
)

BASE_LR = 0.1


def make_optimizer(lr_scale=None):
    groups = [{'params': [torch.nn.Parameter(torch.zeros(2))], 'lr': BASE_LR}]
    if lr_scale is not None:
        groups.append({
            'params': [torch.nn.Parameter(torch.zeros(2))],
            'lr': BASE_LR, 'lr_scale': lr_scale,
        })
    return torch.optim.SGD(groups, lr=BASE_LR)


def lr_curve(sched, steps, group=0):
    values = []
    for t in range(steps):
        sched.step(t)
        values.append(sched.optimizer.param_groups[group]['lr'])
    return values


def test_cosine_endpoints_and_floor():
    opt = make_optimizer()
    s = CosineLRScheduler(opt, t_initial=10, lr_min=1e-3)
    curve = lr_curve(s, 10)
    assert curve[0] == pytest.approx(BASE_LR)
    # halfway point of cosine = midpoint of (base, min)
    assert curve[5] == pytest.approx((BASE_LR + 1e-3) / 2)
    assert min(curve) >= 1e-3
    # past the single allowed cycle -> pinned at lr_min
    s.step(25)
    assert opt.param_groups[0]['lr'] == pytest.approx(1e-3)


def test_warmup_ramp_monotonic():
    for cls in (CosineLRScheduler, TanhLRScheduler, PolyLRScheduler):
        opt = make_optimizer()
        s = cls(opt, t_initial=20, warmup_t=5, warmup_lr_init=1e-5)
        curve = lr_curve(s, 5)
        assert curve[0] == pytest.approx(1e-5)
        assert all(b > a for a, b in zip(curve, curve[1:])), cls.__name__


def test_warmup_prefix_shifts_decay_clock():
    opt_a, opt_b = make_optimizer(), make_optimizer()
    plain = CosineLRScheduler(opt_a, t_initial=10, warmup_t=4, warmup_lr_init=1e-5)
    prefix = CosineLRScheduler(
        opt_b, t_initial=10, warmup_t=4, warmup_lr_init=1e-5, warmup_prefix=True)
    plain.step(4)
    prefix.step(4)
    # with prefix, decay restarts at t-warmup_t=0 -> full base lr
    assert opt_b.param_groups[0]['lr'] == pytest.approx(BASE_LR)
    assert opt_a.param_groups[0]['lr'] < BASE_LR


def test_cycles_restart_and_decay():
    opt = make_optimizer()
    s = CosineLRScheduler(opt, t_initial=8, cycle_limit=3, cycle_decay=0.5)
    curve = lr_curve(s, 24)
    assert curve[8] == pytest.approx(BASE_LR * 0.5)   # second cycle peak
    assert curve[16] == pytest.approx(BASE_LR * 0.25)  # third cycle peak
    assert s.get_cycle_length() == 24


def test_cycle_mul_geometry():
    opt = make_optimizer()
    s = CosineLRScheduler(opt, t_initial=4, cycle_mul=2., cycle_limit=3)
    # spans 4, 8, 16 -> total 28
    assert s.get_cycle_length() == 28
    curve = lr_curve(s, 28)
    assert curve[4] == pytest.approx(BASE_LR)   # restart at t=4
    assert curve[12] == pytest.approx(BASE_LR)  # restart at t=12


def test_k_decay_changes_midpoint():
    base = lr_curve(CosineLRScheduler(make_optimizer(), t_initial=10), 10)
    k = lr_curve(CosineLRScheduler(make_optimizer(), t_initial=10, k_decay=2.0), 10)
    assert k[0] == pytest.approx(base[0])
    # k>1 holds lr higher early in the cycle
    assert k[3] > base[3]


def test_poly_power_one_is_linear():
    curve = lr_curve(PolyLRScheduler(make_optimizer(), t_initial=10, power=1.0), 10)
    diffs = [a - b for a, b in zip(curve, curve[1:])]
    assert all(d == pytest.approx(diffs[0]) for d in diffs)


def test_tanh_monotonic_decay():
    curve = lr_curve(TanhLRScheduler(make_optimizer(), t_initial=12), 12)
    assert all(a >= b for a, b in zip(curve, curve[1:]))
    assert curve[-1] < 0.02 * BASE_LR


def test_step_decay_intervals():
    curve = lr_curve(StepLRScheduler(make_optimizer(), decay_t=4, decay_rate=0.1), 12)
    assert curve[:4] == pytest.approx([BASE_LR] * 4)
    assert curve[4:8] == pytest.approx([BASE_LR * 0.1] * 4)
    assert curve[8:] == pytest.approx([BASE_LR * 0.01] * 4)


def test_multistep_milestones():
    curve = lr_curve(
        MultiStepLRScheduler(make_optimizer(), decay_t=[3, 7], decay_rate=0.1), 10)
    # milestone M applies from the step() call with t = M-1 (end of epoch M-1)
    assert curve[1] == pytest.approx(BASE_LR)
    assert curve[3] == pytest.approx(BASE_LR * 0.1)
    assert curve[7] == pytest.approx(BASE_LR * 0.01)


def test_plateau_decays_on_stall():
    opt = make_optimizer()
    s = PlateauLRScheduler(opt, decay_rate=0.5, patience_t=2, mode='max')
    for t in range(10):
        s.step(t, metric=1.0)  # metric never improves
    assert opt.param_groups[0]['lr'] < BASE_LR


def test_noise_bounded_and_seeded():
    curves = []
    for _ in range(2):
        opt = make_optimizer()
        s = CosineLRScheduler(
            opt, t_initial=50, noise_range_t=0, noise_pct=0.3, noise_seed=11)
        curves.append(lr_curve(s, 30))
    assert curves[0] == curves[1]  # seeded -> deterministic
    clean = lr_curve(CosineLRScheduler(make_optimizer(), t_initial=50), 30)
    for noisy, base in zip(curves[0], clean):
        assert abs(noisy - base) <= abs(base) * 0.3 + 1e-12


def test_lr_scale_applied_per_group():
    opt = make_optimizer(lr_scale=0.25)
    s = CosineLRScheduler(opt, t_initial=10)
    s.step(3)
    g0, g1 = opt.param_groups
    assert g1['lr'] == pytest.approx(g0['lr'] * 0.25)


def test_state_dict_roundtrip():
    opt = make_optimizer()
    s1 = CosineLRScheduler(opt, t_initial=10, warmup_t=3, warmup_lr_init=1e-5)
    s1.step(4)
    state = s1.state_dict()
    s2 = CosineLRScheduler(make_optimizer(), t_initial=99)
    s2.load_state_dict(state)
    assert s2.t_initial == 10 and s2.warmup_t == 3
    s1.step(7)
    s2.step(7)
    assert s1.optimizer.param_groups[0]['lr'] == \
        pytest.approx(s2.optimizer.param_groups[0]['lr'])


def test_step_update_only_when_scheduling_on_updates():
    opt = make_optimizer()
    s = CosineLRScheduler(opt, t_initial=100, t_in_epochs=False)
    before = opt.param_groups[0]['lr']
    s.step(50)  # epoch stepping ignored
    assert opt.param_groups[0]['lr'] == before
    s.step_update(50)
    assert opt.param_groups[0]['lr'] != before


def test_factory_epoch_to_update_conversion():
    opt = make_optimizer()
    sched, epochs = create_scheduler_v2(
        opt, sched='cosine', num_epochs=10, warmup_epochs=2,
        step_on_epochs=False, updates_per_epoch=7)
    assert epochs == 10
    assert sched.t_initial == 70 and sched.warmup_t == 14


def test_factory_cycle_extends_epochs():
    opt = make_optimizer()
    _, epochs = create_scheduler_v2(
        opt, sched='cosine', num_epochs=10, cooldown_epochs=2,
        cycle_limit=3)
    assert epochs == 32
