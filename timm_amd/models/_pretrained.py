"""Pretrained-weight descriptor dataclasses.

Behavioral parity: /root/reference/timm/models/_pretrained.py:11-120.  Field
names/defaults are the serialized interface (they round-trip through hub
config JSON and checkpoint metadata) and therefore match the reference
exactly; only the surrounding machinery differs.
"""
from collections import deque
from dataclasses import asdict, dataclass, field
from typing import Any, Dict, Optional, Tuple, Union

__all__ = ['PretrainedCfg', 'filter_pretrained_cfg', 'DefaultCfg']

# keys identifying where weights come from (strippable for publishing)
_SOURCE_KEYS = ('url', 'file', 'hf_hub_id', 'hf_hub_filename', 'source', 'state_dict')
# null-valued keys that still carry meaning and survive null-filtering
_KEEP_NULL = ('pool_size', 'first_conv', 'classifier')


@dataclass
class PretrainedCfg:
    """One pretrained weight source + the data/head config needed to use it."""

    # where the weights live (first available of url/file/state_dict/hub)
    url: Optional[Union[str, Tuple[str, str]]] = None
    file: Optional[str] = None
    state_dict: Optional[Dict[str, Any]] = None
    hf_hub_id: Optional[str] = None
    hf_hub_filename: Optional[str] = None

    source: Optional[str] = None        # which source was actually used
    architecture: Optional[str] = None  # arch name when not implied by registry
    tag: Optional[str] = None           # pretrained tag of this cfg
    custom_load: bool = False           # arch provides its own load fn

    # input pipeline expectations
    input_size: Tuple[int, int, int] = (3, 224, 224)
    test_input_size: Optional[Tuple[int, int, int]] = None
    min_input_size: Optional[Tuple[int, int, int]] = None
    fixed_input_size: bool = False
    interpolation: str = 'bicubic'
    crop_pct: float = 0.875
    test_crop_pct: Optional[float] = None
    crop_mode: str = 'center'
    mean: Tuple[float, ...] = (0.485, 0.456, 0.406)
    std: Tuple[float, ...] = (0.229, 0.224, 0.225)

    # classifier head + labels
    num_classes: int = 1000
    label_offset: Optional[int] = None
    label_names: Optional[Tuple[str]] = None
    label_descriptions: Optional[Dict[str, str]] = None

    # arch attributes needed when adapting weights (stem/head renames etc.)
    pool_size: Optional[Tuple[int, ...]] = None
    test_pool_size: Optional[Tuple[int, ...]] = None
    first_conv: Optional[Union[str, Tuple[str]]] = None
    classifier: Optional[Union[str, Tuple[str]]] = None

    # provenance / documentation
    license: Optional[str] = None
    description: Optional[str] = None
    origin_url: Optional[str] = None
    paper_name: Optional[str] = None
    paper_ids: Optional[Union[str, Tuple[str]]] = None
    notes: Optional[Tuple[str]] = None

    @property
    def has_weights(self) -> bool:
        return bool(self.url or self.file or self.hf_hub_id or self.state_dict is not None)

    def to_dict(self, remove_source=False, remove_null=True):
        return filter_pretrained_cfg(
            asdict(self), remove_source=remove_source, remove_null=remove_null)


def filter_pretrained_cfg(cfg: Dict[str, Any], remove_source=False, remove_null=True):
    """Drop source keys and/or uninformative nulls from a cfg dict."""
    def _keep(key, value):
        if remove_source and key in _SOURCE_KEYS:
            return False
        if remove_null and value is None and key not in _KEEP_NULL:
            return False
        return True

    return {k: v for k, v in cfg.items() if _keep(k, v)}


@dataclass
class DefaultCfg:
    """The tag set of one architecture; tags[0] resolves the bare arch name."""
    tags: deque = field(default_factory=deque)
    cfgs: Dict[str, PretrainedCfg] = field(default_factory=dict)
    is_pretrained: bool = False  # any cfg has a weight source

    @property
    def default(self) -> PretrainedCfg:
        return self.cfgs[self.tags[0]]

    @property
    def default_with_tag(self) -> Tuple[str, PretrainedCfg]:
        tag = self.tags[0]
        return tag, self.cfgs[tag]
