from .adamw import AdamW

try:
    from ._optim_factory import (
        create_optimizer_v2, create_optimizer, optimizer_kwargs, list_optimizers,
        get_optimizer_class, OptimizerRegistry, OptimInfo,
    )
    from ._param_groups import param_groups_layer_decay, param_groups_weight_decay
except ImportError:
    pass
