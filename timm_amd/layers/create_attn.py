"""CNN attention-module factory (reference `timm/layers/create_attn.py`):
resolve a short name / bool / class to an attention-module constructor.
All constructors take the channel count as their first positional arg."""
import functools

import torch

from .cbam import CbamModule, LightCbamModule
from .conv_self_attn import BottleneckAttn, HaloAttn, LambdaLayer
from .coord_attn import CoordAttn, EfficientLocalAttn, SimpleCoordAttn, StripAttn
from .eca import EcaModule, CecaModule
from .gather_excite import GatherExcite
from .global_context import GlobalContext
from .non_local_attn import NonLocalAttn, BatNonLocalAttn
from .selective_kernel import SelectiveKernel
from .split_attn import SplitAttn
from .squeeze_excite import SEModule, EffectiveSEModule

_ATTN_MAP = {
    # lightweight channel / coarse-spatial gates
    'se': SEModule,
    'ese': EffectiveSEModule,
    'eca': EcaModule,
    'ecam': functools.partial(EcaModule, use_mlp=True),
    'ceca': CecaModule,
    'ge': GatherExcite,
    'gc': GlobalContext,
    'gca': functools.partial(GlobalContext, fuse_add=True, fuse_scale=False),
    'cbam': CbamModule,
    'lcbam': LightCbamModule,
    'coord': CoordAttn,
    'scoord': SimpleCoordAttn,
    'ela': EfficientLocalAttn,
    'strip': StripAttn,
    # heavier attention-like blocks
    'sk': SelectiveKernel,
    'splat': SplitAttn,
    'nl': NonLocalAttn,
    'bat': BatNonLocalAttn,
    # spatial self-attention
    'lambda': LambdaLayer,
    'bottleneck': BottleneckAttn,
    'halo': HaloAttn,
}


def get_attn(attn_type):
    if isinstance(attn_type, torch.nn.Module):
        return attn_type
    if not attn_type:
        return None
    if isinstance(attn_type, str):
        key = attn_type.lower()
        assert key in _ATTN_MAP, f'Invalid attn module ({attn_type})'
        return _ATTN_MAP[key]
    if isinstance(attn_type, bool):
        return SEModule if attn_type else None
    return attn_type  # already a class / partial


def create_attn(attn_type, channels, **kwargs):
    module_cls = get_attn(attn_type)
    if module_cls is None:
        return None
    return module_cls(channels, **kwargs)


# explicit-name alias used by model files
create_attn_layer = create_attn
