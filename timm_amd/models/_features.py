"""Feature-extraction wrappers (reference `timm/models/_features.py`, 483 LoC).

Four strategies turn a classifier into a multi-scale feature backbone:

* `FeatureDictNet` / `FeatureListNet` — rebuild the model as a flat module
  dict, tapping outputs at the registered feature points
* `FeatureHookNet` — leave the model intact and capture taps with forward /
  forward-pre hooks
* `FeatureGetterNet` — delegate to the model's own `forward_intermediates()`

`FeatureInfo` carries the per-tap metadata (channels, reduction, module
path); `feature_take_indices` normalizes user index specs.
"""
from collections import OrderedDict, defaultdict
from copy import deepcopy
from functools import partial
from typing import Dict, List, Optional, Sequence, Set, Tuple, Union

import torch
import torch.nn as nn
from torch.utils.checkpoint import checkpoint

from ..layers import Format, _assert

__all__ = [
    'feature_take_indices', 'FeatureInfo', 'FeatureHooks', 'FeatureDictNet', 'FeatureListNet',
    'FeatureHookNet', 'FeatureGetterNet',
]


def feature_take_indices(
        num_features: int,
        indices: Optional[Union[int, List[int]]] = None,
        as_set: bool = False,
) -> Tuple[List[int], int]:
    """Normalize an index spec to absolute indices + the max index
    (reference `_features.py:28`). `indices`: None = all, int = last n,
    list = absolute/negative positions."""
    if indices is None:
        indices = num_features

    if isinstance(indices, int):
        _assert(0 < indices <= num_features, f'last-n ({indices}) is out of range (1 to {num_features})')
        take_indices = [num_features - indices + i for i in range(indices)]
    else:
        take_indices = []
        for i in indices:
            idx = num_features + i if i < 0 else i
            _assert(0 <= idx < num_features, f'feature index {idx} is out of range (0 to {num_features - 1})')
            take_indices.append(idx)

    if not torch.jit.is_scripting() and as_set:
        return set(take_indices), max(take_indices)
    return take_indices, max(take_indices)


def _out_indices_as_tuple(x: Union[int, Tuple[int, ...]]) -> Tuple[int, ...]:
    if isinstance(x, int):
        return tuple(range(-x, 0))  # int spec = last N taps
    return tuple(x)


OutIndicesT = Union[int, Tuple[int, ...]]


class FeatureInfo:
    """Validated list of feature-tap dicts + the selected out indices."""

    def __init__(self, feature_info: List[Dict], out_indices: OutIndicesT):
        prev_reduction = 1
        for i, fi in enumerate(feature_info):
            # required keys; models may attach extras (e.g. 'stage')
            assert 'num_chs' in fi and fi['num_chs'] > 0
            assert 'reduction' in fi and fi['reduction'] >= prev_reduction
            assert 'module' in fi
            prev_reduction = fi['reduction']
            fi.setdefault('index', i)
        self.out_indices = _out_indices_as_tuple(out_indices)
        self.info = feature_info

    def from_other(self, out_indices: OutIndicesT):
        return FeatureInfo(deepcopy(self.info), out_indices)

    def get(self, key: str, idx: Optional[Union[int, List[int]]] = None):
        """Value(s) of `key` at idx (default: the selected out indices)."""
        if idx is None:
            idx = self.out_indices
        if isinstance(idx, (tuple, list)):
            return [self.info[i][key] for i in idx]
        return self.info[idx][key]

    def get_dicts(self, keys: Optional[List[str]] = None, idx: Optional[Union[int, List[int]]] = None):
        """Info dict(s), optionally projected onto `keys`."""
        def project(i):
            return self.info[i] if keys is None else {k: self.info[i][k] for k in keys}
        if idx is None:
            idx = self.out_indices
        if isinstance(idx, (tuple, list)):
            return [project(i) for i in idx]
        return project(idx)

    def channels(self, idx: Optional[Union[int, List[int]]] = None):
        return self.get('num_chs', idx)

    def reduction(self, idx: Optional[Union[int, List[int]]] = None):
        return self.get('reduction', idx)

    def module_name(self, idx: Optional[Union[int, List[int]]] = None):
        return self.get('module', idx)

    def __getitem__(self, item):
        return self.info[item]

    def __len__(self):
        return len(self.info)


class FeatureHooks:
    """Registers forward/forward-pre hooks and collects their outputs per
    device (reference `_features.py:150`)."""

    def __init__(
            self,
            hooks: Sequence[Union[str, Dict]],
            named_modules: dict,
            out_map: Sequence[Union[int, str]] = None,
            default_hook_type: str = 'forward',
    ):
        self._feature_outputs = defaultdict(OrderedDict)
        self._handles = []
        modules = dict(named_modules)
        for i, spec in enumerate(hooks):
            target = spec if isinstance(spec, str) else spec['module']
            hook_type = default_hook_type
            if isinstance(spec, dict):
                hook_type = spec.get('hook_type', default_hook_type)
            hook_id = out_map[i] if out_map else target
            fn = partial(self._collect, hook_id)
            module = modules[target]
            if hook_type == 'forward_pre':
                self._handles.append(module.register_forward_pre_hook(fn))
            elif hook_type == 'forward':
                self._handles.append(module.register_forward_hook(fn))
            else:
                raise AssertionError('Unsupported hook type')

    def _collect(self, hook_id, *args):
        # last positional arg is the interesting tensor: output for forward
        # hooks, the input tuple for forward-pre hooks
        x = args[-1]
        if isinstance(x, tuple):
            x = x[0]
        self._feature_outputs[x.device][hook_id] = x

    def get_output(self, device) -> Dict[str, torch.tensor]:
        out = self._feature_outputs[device]
        self._feature_outputs[device] = OrderedDict()  # drain
        return out


def _module_list(module, flatten_sequential=False):
    # list (not generator) for torchscript compatibility
    entries = []
    for name, child in module.named_children():
        if flatten_sequential and isinstance(child, nn.Sequential):
            # flatten ONE level of Sequential containers into the parent
            for sub_name, sub_child in child.named_children():
                entries.append((f'{name}_{sub_name}', f'{name}.{sub_name}', sub_child))
        else:
            entries.append((name, name, child))
    return entries


def _get_feature_info(net, out_indices: OutIndicesT):
    feature_info = getattr(net, 'feature_info')
    if isinstance(feature_info, FeatureInfo):
        return feature_info.from_other(out_indices)
    if isinstance(feature_info, (list, tuple)):
        return FeatureInfo(net.feature_info, out_indices)
    raise AssertionError('Provided feature_info is not valid')


def _get_return_layers(feature_info, out_map):
    return {
        name: (out_map[i] if out_map is not None else feature_info.out_indices[i])
        for i, name in enumerate(feature_info.module_name())
    }


class FeatureDictNet(nn.ModuleDict):
    """Rebuilt-model extractor returning {tap_id: tensor}
    (reference `_features.py:230`). Children past the deepest tap are dropped."""

    def __init__(
            self,
            model: nn.Module,
            out_indices: OutIndicesT = (0, 1, 2, 3, 4),
            out_map: Sequence[Union[int, str]] = None,
            output_fmt: str = 'NCHW',
            feature_concat: bool = False,
            flatten_sequential: bool = False,
    ):
        super().__init__()
        self.feature_info = _get_feature_info(model, out_indices)
        self.output_fmt = Format(output_fmt)
        self.concat = feature_concat
        self.grad_checkpointing = False
        self.return_layers = {}

        return_layers = _get_return_layers(self.feature_info, out_map)
        remaining = set(return_layers.keys())
        layers = OrderedDict()
        for new_name, old_name, module in _module_list(model, flatten_sequential=flatten_sequential):
            layers[new_name] = module
            if old_name in remaining:
                self.return_layers[new_name] = str(return_layers[old_name])  # str ids for torchscript
                remaining.remove(old_name)
            if not remaining:
                break
        assert not remaining and len(self.return_layers) == len(return_layers), \
            f'Return layers ({remaining}) are not present in model'
        self.update(layers)

    def set_grad_checkpointing(self, enable: bool = True):
        self.grad_checkpointing = enable

    def _collect(self, x) -> (Dict[str, torch.Tensor]):
        out = OrderedDict()
        for i, (name, module) in enumerate(self.items()):
            if self.grad_checkpointing and not torch.jit.is_scripting():
                # keep the first module (input needs grad) and the last
                # (in-place ops can break under checkpointing) un-checkpointed
                boundary = i == 0 or i == max(len(self) - 1, 0)
                x = module(x) if boundary else checkpoint(module, x)
            else:
                x = module(x)

            if name in self.return_layers:
                out_id = self.return_layers[name]
                if isinstance(x, (tuple, list)) and not self.concat:
                    # multi-output tap: concat on request, else take the first
                    out[out_id] = torch.cat(x, 1) if self.concat else x[0]
                else:
                    out[out_id] = x
        return out

    def forward(self, x) -> Dict[str, torch.Tensor]:
        return self._collect(x)


class FeatureListNet(FeatureDictNet):
    """Same as FeatureDictNet but returns the taps as a list
    (reference `_features.py:315`)."""

    def __init__(
            self,
            model: nn.Module,
            out_indices: OutIndicesT = (0, 1, 2, 3, 4),
            output_fmt: str = 'NCHW',
            feature_concat: bool = False,
            flatten_sequential: bool = False,
    ):
        super().__init__(
            model,
            out_indices=out_indices,
            output_fmt=output_fmt,
            feature_concat=feature_concat,
            flatten_sequential=flatten_sequential,
        )

    def forward(self, x) -> (List[torch.Tensor]):
        return list(self._collect(x).values())


class FeatureHookNet(nn.ModuleDict):
    """Hook-based extractor: the wrapped model runs unmodified (or lightly
    truncated) and taps are captured by hooks (reference `_features.py:348`)."""

    def __init__(
            self,
            model: nn.Module,
            out_indices: OutIndicesT = (0, 1, 2, 3, 4),
            out_map: Optional[Sequence[Union[int, str]]] = None,
            return_dict: bool = False,
            output_fmt: str = 'NCHW',
            no_rewrite: Optional[bool] = None,
            flatten_sequential: bool = False,
            default_hook_type: str = 'forward',
    ):
        super().__init__()
        assert not torch.jit.is_scripting()
        self.feature_info = _get_feature_info(model, out_indices)
        self.return_dict = return_dict
        self.output_fmt = Format(output_fmt)
        self.grad_checkpointing = False
        if no_rewrite is None:
            no_rewrite = not flatten_sequential

        layers = OrderedDict()
        hooks = []
        if no_rewrite:
            if hasattr(model, 'reset_classifier'):
                model.reset_classifier(0)  # head output is never a tap
            layers['body'] = model
            hooks.extend(self.feature_info.get_dicts())
        else:
            remaining = {
                f['module']: f.get('hook_type', default_hook_type)
                for f in self.feature_info.get_dicts()
            }
            for new_name, old_name, module in _module_list(model, flatten_sequential=flatten_sequential):
                layers[new_name] = module
                for fn, _ in module.named_modules(prefix=old_name):
                    if fn in remaining:
                        hooks.append(dict(module=fn, hook_type=remaining.pop(fn)))
                if not remaining:
                    break
            assert not remaining, f'Return layers ({remaining}) are not present in model'
        self.update(layers)
        self.hooks = FeatureHooks(hooks, model.named_modules(), out_map=out_map)

    def set_grad_checkpointing(self, enable: bool = True):
        self.grad_checkpointing = enable

    def forward(self, x):
        for i, (name, module) in enumerate(self.items()):
            if self.grad_checkpointing and not torch.jit.is_scripting():
                boundary = i == 0 or i == max(len(self) - 1, 0)
                x = module(x) if boundary else checkpoint(module, x)
            else:
                x = module(x)
        out = self.hooks.get_output(x.device)
        return out if self.return_dict else list(out.values())


class FeatureGetterNet(nn.ModuleDict):
    """Extractor over the model's own `forward_intermediates()` API
    (reference `_features.py:435`)."""

    def __init__(
            self,
            model: nn.Module,
            out_indices: OutIndicesT = 4,
            out_map: Optional[Sequence[Union[int, str]]] = None,
            return_dict: bool = False,
            output_fmt: str = 'NCHW',
            norm: bool = False,
            prune: bool = True,
    ):
        super().__init__()
        if prune and hasattr(model, 'prune_intermediate_layers'):
            # pruning normalizes the indices, so adopt the returned spec
            out_indices = model.prune_intermediate_layers(
                out_indices,
                prune_norm=not norm,
            )
        self.feature_info = _get_feature_info(model, out_indices)
        self.model = model
        self.out_indices = out_indices
        self.out_map = out_map
        self.return_dict = return_dict
        self.output_fmt = Format(output_fmt)
        self.norm = norm

    def set_grad_checkpointing(self, enable: bool = True):
        self.model.set_grad_checkpointing(enable)

    def forward(self, x):
        return self.model.forward_intermediates(
            x,
            indices=self.out_indices,
            norm=self.norm,
            output_fmt=self.output_fmt,
            intermediates_only=True,
        )
