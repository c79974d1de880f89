"""Optimizer tests (reference style: /root/reference/tests/test_optim.py —
factory lookup, convergence on a simple problem, param-group behavior)."""
import pytest
import torch

from timm_amd.optim import create_optimizer_v2, list_optimizers, get_optimizer_class
from timm_amd.optim._param_groups import param_groups_weight_decay

# every registered family gets a convergence smoke at a workable lr
CONVERGENCE_CASES = [
    ('sgd', 0.1), ('momentum', 0.1), ('sgdw', 0.1), ('sgdp', 0.1),
    ('adam', 0.05), ('adamw', 0.05), ('adamp', 0.05), ('nadamw', 0.05),
    ('radam', 0.05), ('adopt', 0.05), ('adabelief', 0.05),
    ('lamb', 0.05), ('lars', 0.5), ('lion', 0.03),
    ('madgrad', 0.05), ('mars', 0.05), ('adan', 0.05),
    ('laprop', 0.05), ('novograd', 0.05), ('adafactorbv', 0.05),
    ('kron', 0.05), ('adahessian', 0.1), ('muon', 0.1),
    ('rmsproptf', 0.01), ('adafactor', None),
]


def _toy_params():
    torch.manual_seed(9)
    return torch.nn.ParameterList([
        torch.nn.Parameter(torch.randn(8, 10) * 2),
        torch.nn.Parameter(torch.randn(10)),
    ])


@pytest.mark.parametrize('opt_name,lr', CONVERGENCE_CASES)
def test_optimizer_converges(opt_name, lr):
    params = _toy_params()
    kwargs = dict(opt=opt_name)
    if lr is not None:
        kwargs['lr'] = lr
    opt = create_optimizer_v2(params, **kwargs)
    second_order = getattr(opt, 'is_second_order', False)
    initial = None
    for _ in range(150):
        opt.zero_grad()
        loss = sum((p ** 2).sum() for p in params)
        if initial is None:
            initial = loss.item()
        loss.backward(create_graph=second_order)
        opt.step()
    final = sum((p ** 2).sum() for p in params).item()
    assert final < 0.2 * initial, f'{opt_name}: {initial} -> {final}'


def test_list_and_lookup():
    names = list_optimizers()
    assert len(names) > 50
    for required in ('adamw', 'lamb', 'muon', 'kron', 'adahessian', 'sgdp',
                     'adamp', 'laprop', 'novograd', 'adafactorbv'):
        assert required in names, required
        assert get_optimizer_class(required) is not None


def test_adahessian_is_second_order():
    params = _toy_params()
    opt = create_optimizer_v2(params, opt='adahessian', lr=0.1)
    assert opt.is_second_order


def test_no_weight_decay_on_1d_params():
    model = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.LayerNorm(8))
    groups = param_groups_weight_decay(model, weight_decay=0.05)
    assert len(groups) == 2
    no_decay, decay = groups[0], groups[1]
    assert no_decay['weight_decay'] == 0.
    assert decay['weight_decay'] == 0.05
    # all 1d params (biases + LN) in the no-decay group
    assert all(p.ndim <= 1 for p in no_decay['params'])
    assert all(p.ndim > 1 for p in decay['params'])


def test_lookahead_prefix():
    params = _toy_params()
    opt = create_optimizer_v2(params, opt='lookahead_adamw', lr=0.05)
    from timm_amd.optim.lookahead import Lookahead
    assert isinstance(opt, Lookahead)


def test_weight_decay_applied():
    p = torch.nn.Parameter(torch.ones(4, 4))
    opt = create_optimizer_v2([p], opt='adamw', lr=0.1, weight_decay=0.5)
    p.grad = torch.zeros_like(p)
    opt.step()
    assert p.abs().max().item() < 1.0  # decay shrank weights despite zero grad
