"""Model registry: name -> entrypoint / pretrained-cfg bookkeeping.

Behavioral parity: /root/reference/timm/models/_registry.py (decorator
registration at import time, arch.tag naming, wildcard listing with tag
expansion, deprecation shims).

Redesigned as a single encapsulated ``ModelRegistry`` object (the reference
keeps eight parallel module-global dicts); module-level functions delegate to
the default instance so the public API is unchanged.
"""
import fnmatch
import re
import sys
import warnings
from collections import defaultdict, deque
from copy import deepcopy
from dataclasses import replace
from typing import Any, Callable, Dict, List, Optional, Sequence, Set, Tuple, Union

from ._pretrained import DefaultCfg, PretrainedCfg

__all__ = [
    'split_model_name_tag', 'get_arch_name', 'register_model', 'generate_default_cfgs',
    'list_models', 'list_pretrained', 'is_model', 'model_entrypoint', 'list_modules',
    'is_model_in_modules', 'get_pretrained_cfg_value', 'is_model_pretrained',
    'get_pretrained_cfg', 'get_pretrained_cfgs_for_arch', 'register_model_deprecations',
]


def split_model_name_tag(model_name: str, no_tag: str = '') -> Tuple[str, str]:
    """'arch.tag' -> (arch, tag); tag defaults to ``no_tag`` when absent."""
    arch, _, tag = model_name.partition('.')
    return arch, tag or no_tag


def get_arch_name(model_name: str) -> str:
    return split_model_name_tag(model_name)[0]


def _natural_key(s: str) -> List[Union[int, str]]:
    return [int(p) if p.isdigit() else p for p in re.split(r'(\d+)', s.lower())]


class ModelRegistry:
    """All model-name state in one place."""

    def __init__(self):
        self.entrypoints: Dict[str, Callable[..., Any]] = {}
        self.arch_module: Dict[str, str] = {}
        self.module_archs: Dict[str, Set[str]] = defaultdict(set)
        self.pretrained_names: Set[str] = set()  # names (arch or arch.tag) with weights
        self.default_cfgs: Dict[str, DefaultCfg] = {}
        self.pretrained_cfgs: Dict[str, PretrainedCfg] = {}
        self.arch_tagged_names: Dict[str, List[str]] = defaultdict(list)
        self.deprecated: Dict[str, Optional[str]] = {}
        self.module_deprecated: Dict[str, Dict[str, Optional[str]]] = defaultdict(dict)

    # -- registration -------------------------------------------------------
    def add(self, fn: Callable[..., Any]) -> Callable[..., Any]:
        module = sys.modules[fn.__module__]
        module_name = fn.__module__.rsplit('.', 1)[-1]
        arch = fn.__name__

        # expose through the defining module's __all__
        if hasattr(module, '__all__'):
            module.__all__.append(arch)
        else:
            module.__all__ = [arch]

        if arch in self.entrypoints:
            warnings.warn(
                f'Overwriting {arch} in registry with {fn.__module__}.{arch}. This is '
                'because the name being registered conflicts with an existing name. '
                'Please check if this is not expected.',
                stacklevel=2)
        self.entrypoints[arch] = fn
        self.arch_module[arch] = module_name
        self.module_archs[module_name].add(arch)

        cfg = getattr(module, 'default_cfgs', {}).get(arch, None)
        if cfg is not None:
            self._add_cfgs(arch, cfg)
        return fn

    def _add_cfgs(self, arch: str, default_cfg):
        if not isinstance(default_cfg, DefaultCfg):
            # legacy single-dict cfg -> one untagged entry
            assert isinstance(default_cfg, dict)
            default_cfg = DefaultCfg(
                tags=deque(['']), cfgs={'': PretrainedCfg(**default_cfg)})

        for idx, tag in enumerate(default_cfg.tags):
            cfg = default_cfg.cfgs[tag]
            full_name = f'{arch}.{tag}' if tag else arch
            fixups = dict(architecture=arch, tag=tag or None)
            if cfg.hf_hub_id == 'timm/':
                fixups['hf_hub_id'] = 'timm/' + full_name  # hub repo named by arch.tag
            cfg = replace(cfg, **fixups)

            if idx == 0:
                # first tag is the default resolution for the bare arch name
                self.pretrained_cfgs[arch] = cfg
                if cfg.has_weights:
                    self.pretrained_names.add(arch)
            if tag:
                self.pretrained_cfgs[full_name] = cfg
                if cfg.has_weights:
                    self.pretrained_names.add(full_name)
            self.arch_tagged_names[arch].append(full_name)
        self.default_cfgs[arch] = default_cfg

    def add_deprecations(self, module_path: str, mapping: Dict[str, Optional[str]]):
        module = sys.modules[module_path]
        module_name = module_path.rsplit('.', 1)[-1]
        for old_name, new_name in mapping.items():
            if hasattr(module, '__all__'):
                module.__all__.append(old_name)
            target_fn, target_tag = None, ''
            if new_name:
                target_arch, target_tag = split_model_name_tag(new_name)
                target_fn = getattr(module, target_arch)
            shim = _deprecation_shim(old_name, target_fn, target_tag)
            setattr(module, old_name, shim)
            self.entrypoints[old_name] = shim
            self.arch_module[old_name] = module_name
            self.module_archs[module_name].add(old_name)
            self.deprecated[old_name] = new_name
            self.module_deprecated[module_name][old_name] = new_name

    # -- queries ------------------------------------------------------------
    def names(
            self,
            filter: Union[str, List[str]] = '',
            module: Union[str, List[str]] = '',
            pretrained: bool = False,
            exclude_filters: Union[str, List[str]] = '',
            name_matches_cfg: bool = False,
            include_tags: Optional[bool] = None,
    ) -> List[str]:
        if include_tags is None:
            include_tags = pretrained  # tag expansion implied for weight listings

        if not module:
            pool: Set[str] = set(self.entrypoints)
        elif isinstance(module, str):
            pool = set(self.module_archs[module])
        else:
            assert isinstance(module, Sequence)
            pool = set()
            for m in module:
                pool |= self.module_archs[m]
        pool -= self.deprecated.keys()

        include = [filter] if isinstance(filter, str) and filter else list(filter or [])
        exclude = [exclude_filters] if isinstance(exclude_filters, str) and exclude_filters \
            else list(exclude_filters or [])

        if include_tags:
            expanded: Set[str] = set()
            for arch in pool:
                expanded.update(self.arch_tagged_names[arch])
            pool = expanded
            include = [x for f in include for x in _with_tag_wildcard(f)]
            exclude = [x for f in exclude for x in _with_tag_wildcard(f)]

        if include:
            selected: Set[str] = set()
            for pattern in include:
                selected |= set(fnmatch.filter(pool, pattern))
        else:
            selected = pool
        for pattern in exclude:
            selected -= set(fnmatch.filter(selected, pattern))

        if pretrained:
            selected &= self.pretrained_names
        if name_matches_cfg:
            selected &= set(self.pretrained_cfgs)
        return sorted(selected, key=_natural_key)

    def entrypoint(self, model_name: str, module_filter: Optional[str] = None):
        arch = get_arch_name(model_name)
        if module_filter and arch not in self.module_archs.get(module_filter, {}):
            raise RuntimeError(f'Model ({model_name} not found in module {module_filter}.')
        return self.entrypoints[arch]

    def cfg_for(self, model_name: str, allow_unregistered: bool = True):
        if model_name in self.pretrained_cfgs:
            return deepcopy(self.pretrained_cfgs[model_name])
        arch, tag = split_model_name_tag(model_name)
        if arch in self.default_cfgs:
            raise RuntimeError(f'Invalid pretrained tag ({tag}) for {arch}.')
        if allow_unregistered:
            return None
        raise RuntimeError(f'Model architecture ({arch}) has no pretrained cfg registered.')


def _with_tag_wildcard(pattern: str) -> List[str]:
    """Untagged filter 'x' also matches tagged names via 'x.*'."""
    base, tag = split_model_name_tag(pattern)
    if tag:
        return [pattern]
    return [f'{base}.*', pattern]


def _deprecation_shim(old_name: str, target_fn: Optional[Callable], target_tag: str):
    def _shim(pretrained=False, **kwargs):
        assert target_fn is not None, \
            f'Model {old_name} has been removed with no replacement.'
        new_name = f'{target_fn.__name__}.{target_tag}' if target_tag else target_fn.__name__
        warnings.warn(
            f'Mapping deprecated model name {old_name} to current {new_name}.',
            stacklevel=2)
        pretrained_cfg = kwargs.pop('pretrained_cfg', None)
        return target_fn(
            pretrained=pretrained, pretrained_cfg=pretrained_cfg or target_tag, **kwargs)
    return _shim


# the default (and only) registry instance + reference-compatible aliases of
# its internal tables for external pokes
_registry = ModelRegistry()
_model_entrypoints = _registry.entrypoints
_model_to_module = _registry.arch_module
_module_to_models = _registry.module_archs
_model_has_pretrained = _registry.pretrained_names
_model_default_cfgs = _registry.default_cfgs
_model_pretrained_cfgs = _registry.pretrained_cfgs
_model_with_tags = _registry.arch_tagged_names
_deprecated_models = _registry.deprecated
_module_to_deprecated_models = _registry.module_deprecated


def generate_default_cfgs(cfgs: Dict[str, Union[Dict[str, Any], PretrainedCfg]]):
    """Group 'arch.tag' -> cfg mappings into per-arch DefaultCfg objects.

    Default-tag priority: explicit untagged-with-weights first, then
    '*'-suffixed tags, then the first tag with weights.
    """
    out = defaultdict(DefaultCfg)
    explicit_default: Set[str] = set()
    for name, cfg in cfgs.items():
        if isinstance(cfg, dict):
            cfg = PretrainedCfg(**cfg)
        arch, tag = split_model_name_tag(name)
        promote = (cfg.has_weights and not tag) or \
            (tag.endswith('*') and arch not in explicit_default)
        tag = tag.strip('*')

        entry = out[arch]
        if promote:
            entry.tags.appendleft(tag)
            explicit_default.add(arch)
        elif cfg.has_weights and not entry.is_pretrained:
            entry.tags.appendleft(tag)
        else:
            entry.tags.append(tag)
        if cfg.has_weights:
            entry.is_pretrained = True
        entry.cfgs[tag] = cfg
    return out


def register_model(fn: Callable[..., Any]) -> Callable[..., Any]:
    return _registry.add(fn)


def register_model_deprecations(module_name: str, deprecation_map: Dict[str, Optional[str]]):
    _registry.add_deprecations(module_name, deprecation_map)


def list_models(
        filter: Union[str, List[str]] = '',
        module: Union[str, List[str]] = '',
        pretrained: bool = False,
        exclude_filters: Union[str, List[str]] = '',
        name_matches_cfg: bool = False,
        include_tags: Optional[bool] = None,
) -> List[str]:
    """List registered model names matching the wildcard filters."""
    return _registry.names(
        filter=filter,
        module=module,
        pretrained=pretrained,
        exclude_filters=exclude_filters,
        name_matches_cfg=name_matches_cfg,
        include_tags=include_tags,
    )


def list_pretrained(filter: Union[str, List[str]] = '', exclude_filters: str = '') -> List[str]:
    return _registry.names(
        filter=filter, pretrained=True, exclude_filters=exclude_filters, include_tags=True)


def get_deprecated_models(module: str = '') -> Dict[str, str]:
    source = _registry.module_deprecated[module] if module else _registry.deprecated
    return deepcopy(source)


def is_model(model_name: str) -> bool:
    return get_arch_name(model_name) in _registry.entrypoints


def model_entrypoint(model_name: str, module_filter: Optional[str] = None) -> Callable[..., Any]:
    return _registry.entrypoint(model_name, module_filter)


def list_modules() -> List[str]:
    return sorted(_registry.module_archs.keys())


def is_model_in_modules(model_name: str, module_names: Union[Tuple, List, Set]) -> bool:
    assert isinstance(module_names, (tuple, list, set))
    arch = get_arch_name(model_name)
    return any(arch in _registry.module_archs[m] for m in module_names)


def is_model_pretrained(model_name: str) -> bool:
    return model_name in _registry.pretrained_names


def get_pretrained_cfg(model_name: str, allow_unregistered: bool = True) -> Optional[PretrainedCfg]:
    return _registry.cfg_for(model_name, allow_unregistered=allow_unregistered)


def get_pretrained_cfg_value(model_name: str, cfg_key: str) -> Optional[Any]:
    cfg = _registry.cfg_for(model_name, allow_unregistered=False)
    return getattr(cfg, cfg_key, None)


def get_pretrained_cfgs_for_arch(model_name: str) -> Dict[str, PretrainedCfg]:
    arch = get_arch_name(model_name)
    if arch not in _registry.default_cfgs:
        return {}
    dcfg = _registry.default_cfgs[arch]
    return {tag: deepcopy(dcfg.cfgs[tag]) for tag in dcfg.tags}
