from .adabelief import AdaBelief
from .adafactor import Adafactor
from .adamw import AdamW
from .adan import Adan
from .adopt import Adopt
from .lamb import Lamb
from .lars import Lars
from .lion import Lion
from .lookahead import Lookahead
from .madgrad import MADGRAD
from .mars import Mars
from .muon import Muon, AdaMuon, zeropower_via_newtonschulz
from .nadamw import NAdamW
from .radam import RAdam
from .rmsprop_tf import RMSpropTF
from .sgdw import SGDW

from ._optim_factory import (
    create_optimizer, create_optimizer_v2, optimizer_kwargs, list_optimizers,
    get_optimizer_class, get_optimizer_info, OptimizerRegistry, OptimInfo, default_registry,
)
from ._param_groups import param_groups_layer_decay, param_groups_weight_decay, auto_group_layers
