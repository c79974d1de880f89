"""Module-tree traversal, parameter grouping and checkpointing helpers
(reference `timm/models/_manipulate.py`, 346 LoC)."""
import collections.abc
import math
import re
from collections import defaultdict
from itertools import chain
from typing import Any, Callable, Dict, Iterator, Optional, Tuple, Type, Union

import torch
import torch.utils.checkpoint
from torch import nn

__all__ = [
    'model_parameters', 'named_apply', 'named_modules', 'named_modules_with_params',
    'group_with_matcher', 'group_modules', 'group_parameters', 'flatten_modules',
    'checkpoint', 'checkpoint_seq', 'adapt_input_conv',
]


def model_parameters(model: nn.Module, exclude_head: bool = False):
    if exclude_head:
        # positional heuristic: the classifier's weight+bias are the last two params
        return list(model.parameters())[:-2]
    return model.parameters()


def _walk(module: nn.Module, name: str, depth_first: bool, include_root: bool):
    """Shared pre/post-order traversal yielding (qualified name, module)."""
    if include_root and not depth_first:
        yield name, module
    for child_name, child in module.named_children():
        qualified = f'{name}.{child_name}' if name else child_name
        yield from _walk(child, qualified, depth_first, True)
    if include_root and depth_first:
        yield name, module


def named_apply(
        fn: Callable,
        module: nn.Module,
        name: str = '',
        depth_first: bool = True,
        include_root: bool = False,
) -> nn.Module:
    for mod_name, mod in _walk(module, name, depth_first, include_root):
        fn(module=mod, name=mod_name)
    return module


def named_modules(
        module: nn.Module,
        name: str = '',
        depth_first: bool = True,
        include_root: bool = False,
):
    yield from _walk(module, name, depth_first, include_root)


def named_modules_with_params(
        module: nn.Module,
        name: str = '',
        depth_first: bool = True,
        include_root: bool = False,
):
    for mod_name, mod in _walk(module, name, depth_first, include_root):
        if mod._parameters:
            yield mod_name, mod


MATCH_PREV_GROUP = (99999,)


def group_with_matcher(
        named_objects: Iterator[Tuple[str, Any]],
        group_matcher: Union[Dict, Callable],
        return_values: bool = False,
        reverse: bool = False,
):
    """Bucket named params/modules into ordered layer groups (reference
    `_manipulate.py:80`); drives layer-wise LR decay in
    `optim/_param_groups.py`.

    A dict matcher maps group name -> regex (or list of (regex, suffix)
    pairs); regex capture groups become sort ordinals. A callable matcher
    returns the ordinal(s) directly. Names nothing matches sort last
    (typically neck/head). A suffix of MATCH_PREV_GROUP merges the bucket
    into the preceding layer id.
    """
    if isinstance(group_matcher, dict):
        compiled = []
        for ordinal, (_, spec) in enumerate(group_matcher.items()):
            if spec is None:
                continue
            if isinstance(spec, (tuple, list)):
                # each sub-spec is (regex, suffix ordinal tuple)
                compiled += [(re.compile(p), (ordinal,), s) for p, s in spec]
            else:
                compiled += [(re.compile(spec), (ordinal,), None)]
        group_matcher = compiled

    def ordinal_of(name):
        if isinstance(group_matcher, (list, tuple)):
            for regex, prefix, suffix in group_matcher:
                m = regex.match(name)
                if m:
                    parts = (prefix, m.groups(), suffix)
                    return tuple(map(float, chain.from_iterable(filter(None, parts))))
            return (float('inf'),)  # unmatched -> last group
        ordinal = group_matcher(name)
        if not isinstance(ordinal, collections.abc.Iterable):
            return (ordinal,)
        return tuple(ordinal)

    buckets = defaultdict(list)
    for name, value in named_objects:
        buckets[ordinal_of(name)].append(value if return_values else name)

    # collapse ordinal tuples into consecutive integer layer ids
    layer_to_members = defaultdict(list)
    layer_id = -1
    for key in sorted(k for k in buckets.keys() if k is not None):
        if layer_id < 0 or key[-1] != MATCH_PREV_GROUP[0]:
            layer_id += 1
        layer_to_members[layer_id].extend(buckets[key])

    if reverse:
        assert not return_values, 'reverse mapping only sensible for name output'
        return {name: lid for lid, names in layer_to_members.items() for name in names}
    return layer_to_members


def group_parameters(module: nn.Module, group_matcher, return_values: bool = False, reverse: bool = False):
    return group_with_matcher(
        module.named_parameters(), group_matcher, return_values=return_values, reverse=reverse)


def group_modules(module: nn.Module, group_matcher, return_values: bool = False, reverse: bool = False):
    return group_with_matcher(
        named_modules_with_params(module), group_matcher, return_values=return_values, reverse=reverse)


def flatten_modules(
        named_modules: Iterator[Tuple[str, nn.Module]],
        depth: int = 1,
        prefix: Union[str, Tuple[str, ...]] = '',
        module_types: Union[str, Tuple[Type[nn.Module]]] = 'sequential',
):
    """Expand container modules up to `depth` levels, yielding leaf entries."""
    tuple_prefix = isinstance(prefix, tuple)
    if isinstance(module_types, str):
        if module_types == 'container':
            module_types = (nn.Sequential, nn.ModuleList, nn.ModuleDict)
        else:
            module_types = (nn.Sequential,)
    for name, module in named_modules:
        if depth and isinstance(module, module_types):
            yield from flatten_modules(
                module.named_children(),
                depth - 1,
                prefix=(name,) if tuple_prefix else name,
                module_types=module_types,
            )
        elif tuple_prefix:
            yield prefix + (name,), module
        else:
            yield (f'{prefix}.{name}' if prefix else name), module


def checkpoint(function, *args, use_reentrant: Optional[bool] = None, **kwargs):
    """torch checkpoint with the session-wide reentrant default (reference `:191`)."""
    if use_reentrant is None:
        from ..layers.config import use_reentrant_ckpt
        use_reentrant = use_reentrant_ckpt()
    return torch.utils.checkpoint.checkpoint(function, *args, use_reentrant=use_reentrant, **kwargs)


def checkpoint_seq(
        functions,
        x,
        every: int = 1,
        flatten: bool = False,
        skip_last: bool = False,
        use_reentrant: Optional[bool] = None,
):
    """Activation-checkpoint a module sequence in segments of `every`
    (reference `_manipulate.py:213-287`)."""
    if use_reentrant is None:
        from ..layers.config import use_reentrant_ckpt
        use_reentrant = use_reentrant_ckpt()

    def segment(start, end):
        def forward(_x):
            for idx in range(start, end + 1):
                _x = functions[idx](_x)
            return _x
        return forward

    if isinstance(functions, torch.nn.Sequential):
        functions = functions.children()
    if flatten:
        functions = chain.from_iterable(functions)
    if not isinstance(functions, (tuple, list)):
        functions = tuple(functions)

    num_checkpointed = len(functions) - 1 if skip_last else len(functions)
    end = -1
    for start in range(0, num_checkpointed, every):
        end = min(start + every - 1, num_checkpointed - 1)
        x = torch.utils.checkpoint.checkpoint(segment(start, end), x, use_reentrant=use_reentrant)
    if skip_last:
        return segment(end + 1, len(functions) - 1)(x)
    return x


def adapt_input_conv(in_chans: int, conv_weight: torch.Tensor) -> torch.Tensor:
    """Reshape a pretrained stem conv for a different input channel count
    (reference `_manipulate.py:289`): grayscale sums the RGB filters,
    in_chans > 3 tiles and rescales them."""
    conv_type = conv_weight.dtype
    conv_weight = conv_weight.float()  # fp16 weights can't .sum() on CPU
    O, I, J, K = conv_weight.shape
    if in_chans == 1:
        if I > 3:
            # space2depth stems carry I = 3 * s2d_factor
            assert conv_weight.shape[1] % 3 == 0
            conv_weight = conv_weight.reshape(O, I // 3, 3, J, K).sum(dim=2)
        else:
            conv_weight = conv_weight.sum(dim=1, keepdim=True)
    elif in_chans != 3:
        if I != 3:
            raise NotImplementedError('Weight format not supported by conversion.')
        repeat = int(math.ceil(in_chans / 3))
        conv_weight = conv_weight.repeat(1, repeat, 1, 1)[:, :in_chans, :, :]
        conv_weight *= 3 / float(in_chans)
    return conv_weight.to(conv_type)
