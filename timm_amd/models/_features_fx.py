"""torch.fx feature-graph extraction (reference `timm/models/_features_fx.py:38,67`)."""
from typing import Callable, Dict, List, Optional, Sequence, Set, Tuple, Union

import torch
from torch import nn

from ._features import _get_feature_info, _get_return_layers

try:
    # NOTE we wrap torchvision fns to use timm leaf / no trace definitions
    from torchvision.models.feature_extraction import create_feature_extractor as _create_feature_extractor
    from torchvision.models.feature_extraction import get_graph_node_names as _get_graph_node_names
    has_fx_feature_extraction = True
except ImportError:
    has_fx_feature_extraction = False

__all__ = [
    'register_notrace_module', 'is_notrace_module', 'get_notrace_modules',
    'register_notrace_function', 'is_notrace_function', 'get_notrace_functions',
    'create_feature_extractor', 'get_graph_node_names', 'FeatureGraphNet', 'GraphExtractNet',
]

# modules to treat as leafs when tracing
_leaf_modules = set()

# functions to autowrap (treat as leaves, don't trace into)
_autowrap_functions = set()


def register_notrace_module(module: nn.Module):
    """Decorator for modules that cannot be traced through (dynamic control flow etc.)."""
    _leaf_modules.add(module)
    return module


def is_notrace_module(module: nn.Module):
    return module in _leaf_modules


def get_notrace_modules():
    return list(_leaf_modules)


def register_notrace_function(func: Callable):
    """Decorator for functions which ought not to be traced through."""
    _autowrap_functions.add(func)
    return func


def is_notrace_function(func: Callable):
    return func in _autowrap_functions


def get_notrace_functions():
    return list(_autowrap_functions)


def get_graph_node_names(model: nn.Module) -> Tuple[List[str], List[str]]:
    if not has_fx_feature_extraction:
        raise RuntimeError('torchvision is required for FX feature extraction')
    return _get_graph_node_names(
        model,
        tracer_kwargs={'leaf_modules': list(_leaf_modules), 'autowrap_functions': list(_autowrap_functions)},
    )


def create_feature_extractor(model: nn.Module, return_nodes: Union[Dict[str, str], List[str]]):
    if not has_fx_feature_extraction:
        raise RuntimeError('torchvision is required for FX feature extraction')
    return _create_feature_extractor(
        model, return_nodes,
        tracer_kwargs={'leaf_modules': list(_leaf_modules), 'autowrap_functions': list(_autowrap_functions)},
    )


class FeatureGraphNet(nn.Module):
    """A FX Graph based feature extractor (reference `_features_fx.py:38`)."""
    return_dict: torch.jit.Final[bool]

    def __init__(
            self,
            model: nn.Module,
            out_indices: Tuple[int, ...],
            out_map: Optional[Dict] = None,
            output_fmt: str = 'NCHW',
            return_dict: bool = False,
    ):
        super().__init__()
        self.feature_info = _get_feature_info(model, out_indices)
        if out_map is not None:
            assert len(out_map) == len(out_indices)
        from ..layers import Format
        self.output_fmt = Format(output_fmt)
        return_nodes = _get_return_layers(self.feature_info, out_map)
        return_nodes = {k: str(v) for k, v in return_nodes.items()}
        self.graph_module = create_feature_extractor(model, return_nodes)
        self.return_dict = return_dict

    def forward(self, x):
        out = self.graph_module(x)
        if self.return_dict:
            return out
        return list(out.values())


class GraphExtractNet(nn.Module):
    """A standalone feature extraction wrapper that maps dict -> list or single tensor."""

    def __init__(
            self,
            model: nn.Module,
            return_nodes: Union[Dict[str, str], List[str]],
            squeeze_out: bool = True,
            return_dict: bool = False,
    ):
        super().__init__()
        self.squeeze_out = squeeze_out
        self.graph_module = create_feature_extractor(model, return_nodes)
        self.return_dict = return_dict

    def forward(self, x) -> Union[List[torch.Tensor], torch.Tensor]:
        out = self.graph_module(x)
        if self.return_dict:
            return out
        out = list(out.values())
        return out[0] if self.squeeze_out and len(out) == 1 else out
