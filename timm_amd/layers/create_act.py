"""Activation factory (reference `timm/layers/create_act.py`)."""
from typing import Callable, Optional, Type, Union

from torch import nn

from .activations import *

_ACT_FN = dict(
    silu=F.silu,
    swish=swish,
    mish=mish,
    relu=F.relu,
    relu6=F.relu6,
    leaky_relu=F.leaky_relu,
    elu=F.elu,
    celu=F.celu,
    selu=F.selu,
    gelu=gelu,
    gelu_tanh=lambda x, inplace=False: F.gelu(x, approximate='tanh'),
    quick_gelu=quick_gelu,
    sigmoid=sigmoid,
    tanh=lambda x, inplace=False: x.tanh(),
    hard_sigmoid=hard_sigmoid,
    hard_swish=hard_swish,
    hard_mish=hard_mish,
    identity=lambda x, inplace=False: x,
)

_ACT_LAYER = dict(
    silu=nn.SiLU,
    swish=nn.SiLU,
    mish=Mish,
    relu=nn.ReLU,
    relu6=nn.ReLU6,
    leaky_relu=nn.LeakyReLU,
    elu=nn.ELU,
    prelu=PReLU,
    celu=nn.CELU,
    selu=nn.SELU,
    gelu=GELU,
    gelu_tanh=GELUTanh,
    quick_gelu=QuickGELU,
    sigmoid=Sigmoid,
    tanh=Tanh,
    hard_sigmoid=HardSigmoid,
    hard_swish=HardSwish,
    hard_mish=HardMish,
    identity=nn.Identity,
)

# names usable by ops.bias_act fused epilogue
_FUSIBLE = {'gelu', 'gelu_tanh', 'silu', 'swish', 'relu', 'identity', 'quick_gelu'}


def fusible_act_name(act_layer) -> Optional[str]:
    """Return ops.bias_act name for an act layer class/instance/name, or None."""
    if act_layer is None:
        return 'identity'
    if isinstance(act_layer, str):
        name = act_layer
    else:
        cls = act_layer if isinstance(act_layer, type) else type(act_layer)
        for name, c in _ACT_LAYER.items():
            if c is cls:
                break
        else:
            if cls is nn.GELU:
                name = 'gelu'
            elif cls is nn.SiLU:
                name = 'silu'
            elif cls is nn.ReLU:
                name = 'relu'
            elif cls is nn.Identity:
                name = 'identity'
            else:
                return None
    if name == 'swish':
        name = 'silu'
    return name if name in _FUSIBLE else None


def get_act_fn(name: Union[Callable, str, None] = 'relu'):
    if name is None:
        return None
    if callable(name):
        return name
    name = name.lower()
    if not name:
        return None
    return _ACT_FN[name]


def get_act_layer(name: Union[Type[nn.Module], str, None] = 'relu'):
    if name is None:
        return None
    if not isinstance(name, str):
        return name
    if not name:
        return None
    name = name.lower()
    return _ACT_LAYER[name]


def create_act_layer(name, inplace=None, **kwargs):
    act_layer = get_act_layer(name)
    if act_layer is None:
        return None
    if inplace is None:
        return act_layer(**kwargs)
    try:
        return act_layer(inplace=inplace, **kwargs)
    except TypeError:
        return act_layer(**kwargs)
