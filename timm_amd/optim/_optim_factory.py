"""Optimizer factory + registry (reference `timm/optim/_optim_factory.py`).

`OptimizerRegistry` (:82), `create_optimizer_v2` (:1199), `list_optimizers`
(:1102), `get_optimizer_class` (:1162); registration groups for
sgd/adam/lamb-lars/cautious/etc (:351-1090).  Name parsing supports the
`lookahead_` prefix and `c`-prefixed cautious variants.
"""
import logging
from dataclasses import dataclass, field
from functools import partial
from typing import Any, Callable, Dict, List, Optional, Set, Tuple, Type, Union

import torch
import torch.nn as nn
import torch.optim

from ._param_groups import param_groups_layer_decay, param_groups_weight_decay
from .adabelief import AdaBelief
from .adafactor import Adafactor
from .adafactor_bv import AdafactorBigVision
from .adahessian import Adahessian
from .adamp import AdamP
from .adamw import AdamW
from .adan import Adan
from .adopt import Adopt
from .kron import Kron
from .lamb import Lamb
from .laprop import LaProp
from .lars import Lars
from .lion import Lion
from .lookahead import Lookahead
from .madgrad import MADGRAD
from .mars import Mars
from .muon import Muon, AdaMuon
from .nadamw import NAdamW
from .nvnovograd import NvNovoGrad
from .radam import RAdam
from .rmsprop_tf import RMSpropTF
from .sgdp import SGDP
from .sgdw import SGDW

_logger = logging.getLogger(__name__)

OptimType = Type[torch.optim.Optimizer]


def _default_weight_decay_exclude(model: nn.Module) -> Set[str]:
    if hasattr(model, 'no_weight_decay'):
        return set(model.no_weight_decay())
    return set()


@dataclass
class OptimInfo:
    """Immutable metadata about an optimizer (reference `_optim_factory.py:58`)."""
    name: str
    opt_class: Union[str, OptimType]
    description: str = ''
    has_eps: bool = True
    has_momentum: bool = False
    has_betas: bool = False
    num_betas: int = 2
    second_order: bool = False
    defaults: Optional[Dict[str, Any]] = None


class OptimizerRegistry:
    """Registry managing optimizer configurations and instantiation
    (reference `_optim_factory.py:82`)."""

    def __init__(self):
        self._optimizers: Dict[str, OptimInfo] = {}
        self._foreach_defaults: Set[str] = {'lion', }

    def register(self, info: OptimInfo) -> None:
        name = info.name.lower()
        if name in self._optimizers:
            _logger.warning(f'Optimizer {name} already registered, overwriting')
        self._optimizers[name] = info

    def register_alias(self, alias: str, target: str) -> None:
        target = target.lower()
        if target not in self._optimizers:
            raise KeyError(f'Optimizer {target} not found')
        self._optimizers[alias.lower()] = self._optimizers[target]

    def list_optimizers(
            self,
            filter: Union[str, List[str]] = '',
            exclude_filters: Optional[List[str]] = None,
            with_description: bool = False,
    ) -> List[Union[str, Tuple[str, str]]]:
        import fnmatch
        names = sorted(self._optimizers.keys())
        if filter:
            if isinstance(filter, str):
                filter = [filter]
            filtered = set()
            for f in filter:
                filtered.update(fnmatch.filter(names, f))
            names = sorted(filtered)
        if exclude_filters:
            for xf in exclude_filters:
                names = [n for n in names if not fnmatch.fnmatch(n, xf)]
        if with_description:
            return [(name, self._optimizers[name].description) for name in names]
        return names

    def get_optimizer_info(self, name: str) -> OptimInfo:
        name = name.lower()
        if name not in self._optimizers:
            raise ValueError(f'Optimizer {name} not found in registry')
        return self._optimizers[name]

    def get_optimizer_class(self, name_or_info, bind_defaults: bool = True):
        if isinstance(name_or_info, str):
            opt_info = self.get_optimizer_info(name_or_info)
        else:
            assert isinstance(name_or_info, OptimInfo)
            opt_info = name_or_info

        opt_class = opt_info.opt_class
        if isinstance(opt_class, str):
            raise ValueError(f'Optimizer {opt_info.name} class unavailable: {opt_class}')

        # Return class or partial with defaults
        if bind_defaults and opt_info.defaults:
            opt_class = partial(opt_class, **opt_info.defaults)

        return opt_class

    def create_optimizer(
            self,
            model_or_params: Union[nn.Module, Any],
            opt: str,
            lr: Optional[float] = None,
            weight_decay: float = 0.,
            momentum: float = 0.9,
            foreach: Optional[bool] = None,
            weight_decay_exclude_1d: bool = True,
            layer_decay: Optional[float] = None,
            layer_decay_min_scale: Optional[float] = None,
            layer_decay_no_opt_scale: Optional[float] = None,
            param_group_fn: Optional[Callable] = None,
            **kwargs: Any,
    ) -> torch.optim.Optimizer:
        """Create an optimizer instance (reference `_optim_factory.py:228-351`)."""
        if isinstance(model_or_params, nn.Module):
            # Extract parameters from a nn.Module, build param groups w/ weight-decay and/or layer-decay applied
            no_weight_decay = _default_weight_decay_exclude(model_or_params)

            if param_group_fn:
                # run custom fn to generate param groups from nn.Module
                params = param_group_fn(model_or_params)
            elif layer_decay is not None:
                params = param_groups_layer_decay(
                    model_or_params,
                    weight_decay=weight_decay,
                    layer_decay=layer_decay,
                    no_weight_decay_list=no_weight_decay,
                    weight_decay_exclude_1d=weight_decay_exclude_1d,
                    min_scale=layer_decay_min_scale or 0.,
                    no_opt_scale=layer_decay_no_opt_scale,
                )
                weight_decay = 0.
            elif weight_decay and weight_decay_exclude_1d:
                params = param_groups_weight_decay(
                    model_or_params,
                    weight_decay=weight_decay,
                    no_weight_decay_list=no_weight_decay,
                )
                weight_decay = 0.
            else:
                params = model_or_params.parameters()
        else:
            # pass parameters / parameter groups through
            params = model_or_params

        # Parse optimizer name
        opt_split = opt.lower().split('_')
        opt_name = opt_split[-1]
        use_lookahead = opt_split[0] == 'lookahead' if len(opt_split) > 1 else False

        opt_info = self.get_optimizer_info(opt_name)

        # Build optimizer arguments
        opt_args: Dict[str, Any] = {'weight_decay': weight_decay, **kwargs}

        # Add LR to args, if None optimizer default is used, some optimizers manage LR internally if None.
        if lr is not None:
            opt_args['lr'] = lr

        # Apply optimizer-specific settings
        if opt_info.defaults:
            for k, v in opt_info.defaults.items():
                opt_args.setdefault(k, v)

        # timm has always defaulted momentum to 0.9 if optimizer supports momentum, keep for backward compat.
        if opt_info.has_momentum:
            opt_args.setdefault('momentum', momentum)

        # Remove commonly used kwargs that aren't always supported
        if not opt_info.has_eps:
            opt_args.pop('eps', None)
        if not opt_info.has_betas:
            opt_args.pop('betas', None)

        if foreach is not None:
            # Explicitly activate or deactivate multi-tensor foreach impl.
            # Not all optimizers support this, and those that do usually default to using
            # multi-tensor impl if foreach is left as default 'None' and can be enabled.
            try:
                opt_class = opt_info.opt_class if not isinstance(opt_info.opt_class, str) else None
                if opt_class is not None and 'foreach' in opt_class.__init__.__code__.co_varnames:
                    opt_args.setdefault('foreach', foreach)
            except AttributeError:
                pass

        # Create optimizer
        opt_class = self.get_optimizer_class(opt_info, bind_defaults=False)
        optimizer = opt_class(params, **opt_args)

        # Apply Lookahead if requested
        if use_lookahead:
            optimizer = Lookahead(optimizer)

        return optimizer


def _register_sgd_variants(registry: OptimizerRegistry) -> None:
    sgd_optimizers = [
        OptimInfo(name='sgd', opt_class=torch.optim.SGD, description='torch.optim SGD (L2 wd) w/ Nesterov',
                  has_eps=False, has_momentum=True, defaults={'nesterov': True}),
        OptimInfo(name='momentum', opt_class=torch.optim.SGD, description='torch.optim SGD w/ classical momentum',
                  has_eps=False, has_momentum=True, defaults={'nesterov': False}),
        OptimInfo(name='sgdp', opt_class=SGDP, description='SGD with scale-invariance projection',
                  has_eps=False, has_momentum=True, defaults={'nesterov': True}),
        OptimInfo(name='sgdw', opt_class=SGDW, description='SGD with decoupled weight decay and Nesterov momentum',
                  has_eps=False, has_momentum=True, defaults={'nesterov': True}),
        OptimInfo(name='csgdw', opt_class=SGDW, description='Cautious SGD with decoupled weight decay',
                  has_eps=False, has_momentum=True, defaults={'nesterov': True, 'caution': True}),
    ]
    for opt in sgd_optimizers:
        registry.register(opt)


def _register_adam_variants(registry: OptimizerRegistry) -> None:
    adam_optimizers = [
        OptimInfo(name='adam', opt_class=torch.optim.Adam, description='torch.optim.Adam', has_betas=True),
        OptimInfo(name='adamw', opt_class=AdamW,
                  description='Adam with decoupled weight decay — fused multi-tensor HIP step', has_betas=True),
        OptimInfo(name='cadamw', opt_class=AdamW, description='Cautious AdamW',
                  has_betas=True, defaults={'caution': True}),
        OptimInfo(name='adamwlegacy', opt_class=AdamW, description='AdamW (alias)', has_betas=True),
        OptimInfo(name='adamax', opt_class=torch.optim.Adamax, description='torch.optim.Adamax', has_betas=True),
        OptimInfo(name='nadam', opt_class=torch.optim.NAdam, description='torch.optim.NAdam', has_betas=True),
        OptimInfo(name='nadamw', opt_class=NAdamW,
                  description='NAdamW - AdamW with Nesterov momentum', has_betas=True),
        OptimInfo(name='cnadamw', opt_class=NAdamW, description='Cautious NAdamW',
                  has_betas=True, defaults={'caution': True}),
        OptimInfo(name='radam', opt_class=RAdam, description='Rectified Adam', has_betas=True),
        OptimInfo(name='cradam', opt_class=RAdam, description='Cautious RAdam',
                  has_betas=True, defaults={'caution': True}),
        OptimInfo(name='adopt', opt_class=Adopt, description='ADOPT - modified Adam'),
        OptimInfo(name='adoptw', opt_class=Adopt, description='ADOPT with decoupled weight decay',
                  defaults={'decoupled': True}),
        OptimInfo(name='cadopt', opt_class=Adopt, description='Cautious ADOPT', defaults={'caution': True}),
        OptimInfo(name='cadoptw', opt_class=Adopt, description='Cautious ADOPT w/ decoupled decay',
                  defaults={'decoupled': True, 'caution': True}),
        OptimInfo(name='adabelief', opt_class=AdaBelief, description='AdaBelief',
                  has_betas=True, defaults={'rectify': False}),
        OptimInfo(name='radabelief', opt_class=AdaBelief, description='Rectified AdaBelief',
                  has_betas=True, defaults={'rectify': True}),
    ]
    for opt in adam_optimizers:
        registry.register(opt)


def _register_lamb_lars(registry: OptimizerRegistry) -> None:
    lamb_lars_optimizers = [
        OptimInfo(name='lamb', opt_class=Lamb, description='LAMB - layer-wise adaptive moments', has_betas=True),
        OptimInfo(name='lambc', opt_class=Lamb, description='LAMB w/ trust-ratio clipping',
                  has_betas=True, defaults={'trust_clip': True}),
        OptimInfo(name='lambw', opt_class=Lamb, description='LAMB w/ decoupled weight decay',
                  has_betas=True, defaults={'decoupled_decay': True}),
        OptimInfo(name='clamb', opt_class=Lamb, description='Cautious LAMB',
                  has_betas=True, defaults={'caution': True}),
        OptimInfo(name='clambw', opt_class=Lamb, description='Cautious LAMB w/ decoupled decay',
                  has_betas=True, defaults={'caution': True, 'decoupled_decay': True}),
        OptimInfo(name='lars', opt_class=Lars, description='LARS', has_momentum=True),
        OptimInfo(name='larc', opt_class=Lars, description='LARS w/ trust-ratio clipping (LARC)',
                  has_momentum=True, defaults={'trust_clip': True}),
        OptimInfo(name='nlars', opt_class=Lars, description='LARS w/ Nesterov momentum',
                  has_momentum=True, defaults={'nesterov': True}),
        OptimInfo(name='nlarc', opt_class=Lars, description='LARC w/ Nesterov momentum',
                  has_momentum=True, defaults={'nesterov': True, 'trust_clip': True}),
    ]
    for opt in lamb_lars_optimizers:
        registry.register(opt)


def _register_other_optimizers(registry: OptimizerRegistry) -> None:
    other_optimizers = [
        OptimInfo(name='adafactor', opt_class=Adafactor, description='Memory-efficient Adafactor'),
        OptimInfo(name='adan', opt_class=Adan, description='Adaptive Nesterov momentum',
                  has_betas=True, num_betas=3),
        OptimInfo(name='adanw', opt_class=Adan, description='Adan w/ decoupled weight decay',
                  has_betas=True, num_betas=3, defaults={'no_prox': True}),
        OptimInfo(name='lion', opt_class=Lion, description='Evolved sign momentum',
                  has_eps=False, has_betas=True),
        OptimInfo(name='clion', opt_class=Lion, description='Cautious Lion',
                  has_eps=False, has_betas=True, defaults={'caution': True}),
        OptimInfo(name='madgrad', opt_class=MADGRAD, description='Momentumized dual-averaged gradient',
                  has_momentum=True),
        OptimInfo(name='madgradw', opt_class=MADGRAD, description='MADGRAD w/ decoupled weight decay',
                  has_momentum=True, defaults={'decoupled_decay': True}),
        OptimInfo(name='mars', opt_class=Mars, description='Variance reduction (MARS)', has_betas=True),
        OptimInfo(name='cmars', opt_class=Mars, description='Cautious MARS',
                  has_betas=True, defaults={'caution': True}),
        OptimInfo(name='muon', opt_class=Muon,
                  description='MomentUm Orthogonalized by Newton-Schulz (bf16 NS on MFMA)',
                  has_eps=False, has_momentum=True),
        OptimInfo(name='adamuon', opt_class=AdaMuon, description='AdaMuon variant',
                  has_eps=False, has_momentum=True),
        OptimInfo(name='cmuon', opt_class=Muon, description='Cautious Muon',
                  has_eps=False, has_momentum=True, defaults={'caution': True}),
        OptimInfo(name='rmsprop', opt_class=torch.optim.RMSprop, description='torch.optim RMSProp',
                  has_momentum=True, defaults={'alpha': 0.9}),
        OptimInfo(name='rmsproptf', opt_class=RMSpropTF, description='TF-style RMSProp',
                  has_momentum=True, defaults={'alpha': 0.9}),
        OptimInfo(name='crmsproptf', opt_class=RMSpropTF, description='Cautious TF RMSProp',
                  has_momentum=True, defaults={'alpha': 0.9, 'caution': True}),
        OptimInfo(name='adafactorbv', opt_class=AdafactorBigVision,
                  description='Big-Vision Adafactor (factored 2nd moments, step-scheduled beta2)'),
        OptimInfo(name='cadafactorbv', opt_class=AdafactorBigVision,
                  description='Cautious Big-Vision Adafactor', defaults={'caution': True}),
        OptimInfo(name='adahessian', opt_class=Adahessian,
                  description='Second-order optimizer w/ Hutchinson Hessian diagonal',
                  second_order=True, has_betas=True),
        OptimInfo(name='adamp', opt_class=AdamP,
                  description='Adam with scale-invariance projection', has_betas=True),
        OptimInfo(name='cadamp', opt_class=AdamP,
                  description='Cautious AdamP (projection, cautious mask n/a -> alias)',
                  has_betas=True),
        OptimInfo(name='kron', opt_class=Kron,
                  description='PSGD-Kron: Kronecker-factored preconditioned SGD',
                  has_momentum=True, has_eps=False),
        OptimInfo(name='kronw', opt_class=Kron,
                  description='PSGD-Kron w/ decoupled weight decay',
                  has_momentum=True, has_eps=False),
        OptimInfo(name='ckron', opt_class=Kron, description='Cautious PSGD-Kron',
                  has_momentum=True, has_eps=False, defaults={'caution': True}),
        OptimInfo(name='laprop', opt_class=LaProp,
                  description='LaProp: decoupled momentum and adaptivity', has_betas=True),
        OptimInfo(name='claprop', opt_class=LaProp, description='Cautious LaProp',
                  has_betas=True, defaults={'caution': True}),
        OptimInfo(name='novograd', opt_class=NvNovoGrad,
                  description='NovoGrad: layer-wise second moments', has_betas=True),
        OptimInfo(name='nvnovograd', opt_class=NvNovoGrad,
                  description='NovoGrad (alias)', has_betas=True),
        OptimInfo(name='adadelta', opt_class=torch.optim.Adadelta, description='torch.optim.Adadelta'),
        OptimInfo(name='adagrad', opt_class=torch.optim.Adagrad, description='torch.optim.Adagrad',
                  defaults={'eps': 1e-8}),
    ]
    for opt in other_optimizers:
        registry.register(opt)


def _register_default_optimizers() -> OptimizerRegistry:
    registry = OptimizerRegistry()
    _register_sgd_variants(registry)
    _register_adam_variants(registry)
    _register_lamb_lars(registry)
    _register_other_optimizers(registry)

    # Register aliases
    registry.register_alias('nesterov', 'sgd')
    registry.register_alias('nesterovw', 'sgdw')

    return registry


# Global registry instance
default_registry = _register_default_optimizers()


def list_optimizers(
        filter: Union[str, List[str]] = '',
        exclude_filters: Optional[List[str]] = None,
        with_description: bool = False,
):
    """List available optimizer names (reference `_optim_factory.py:1102`)."""
    return default_registry.list_optimizers(filter, exclude_filters, with_description)


def get_optimizer_info(name: str) -> OptimInfo:
    return default_registry.get_optimizer_info(name)


def get_optimizer_class(name_or_info, bind_defaults: bool = True):
    """Get optimizer class by name (reference `_optim_factory.py:1162`)."""
    return default_registry.get_optimizer_class(name_or_info, bind_defaults=bind_defaults)


def create_optimizer_v2(
        model_or_params: Union[nn.Module, Any],
        opt: str = 'sgd',
        lr: Optional[float] = None,
        weight_decay: float = 0.,
        momentum: float = 0.9,
        foreach: Optional[bool] = None,
        filter_bias_and_bn: bool = True,
        layer_decay: Optional[float] = None,
        layer_decay_min_scale: float = 0.,
        layer_decay_no_opt_scale: Optional[float] = None,
        param_group_fn: Optional[Callable] = None,
        **kwargs: Any,
) -> torch.optim.Optimizer:
    """Create an optimizer (reference `_optim_factory.py:1199`)."""
    return default_registry.create_optimizer(
        model_or_params,
        opt=opt,
        lr=lr,
        weight_decay=weight_decay,
        momentum=momentum,
        foreach=foreach,
        weight_decay_exclude_1d=filter_bias_and_bn,
        layer_decay=layer_decay,
        layer_decay_min_scale=layer_decay_min_scale,
        layer_decay_no_opt_scale=layer_decay_no_opt_scale,
        param_group_fn=param_group_fn,
        **kwargs,
    )


def optimizer_kwargs(cfg):
    """cfg/argparse to kwargs helper: convert optimizer args in argparse/cfg-like object to kwargs."""
    kwargs = dict(
        opt=cfg.opt,
        lr=cfg.lr,
        weight_decay=cfg.weight_decay,
        momentum=cfg.momentum,
    )
    if getattr(cfg, 'opt_eps', None) is not None:
        kwargs['eps'] = cfg.opt_eps
    if getattr(cfg, 'opt_betas', None) is not None:
        kwargs['betas'] = cfg.opt_betas
    if getattr(cfg, 'layer_decay', None) is not None:
        kwargs['layer_decay'] = cfg.layer_decay
    if getattr(cfg, 'layer_decay_min_scale', None) is not None:
        kwargs['layer_decay_min_scale'] = cfg.layer_decay_min_scale
    if getattr(cfg, 'layer_decay_no_opt_scale', None) is not None:
        kwargs['layer_decay_no_opt_scale'] = cfg.layer_decay_no_opt_scale
    if getattr(cfg, 'opt_args', None) is not None:
        kwargs.update(cfg.opt_args)
    if getattr(cfg, 'opt_foreach', None) is not None:
        kwargs['foreach'] = cfg.opt_foreach
    return kwargs


def create_optimizer(args, model, filter_bias_and_bn=True):
    """Legacy optimizer factory for backwards compat."""
    return create_optimizer_v2(
        model,
        **optimizer_kwargs(cfg=args),
        filter_bias_and_bn=filter_bias_and_bn,
    )
