"""ImageNet synset metadata provider (reference `timm/data/imagenet_info.py`).

Synset lists and lemma/definition tables ship as package data under
`_info/`; the active subset is inferred from the classifier width when not
given explicitly.
"""
import csv
import os
import pkgutil
import re
from typing import Dict, List, Optional, Union

from .dataset_info import DatasetInfo

# classifier width -> subset name (unambiguous so far)
_NUM_CLASSES_TO_SUBSET = {
    1000: 'imagenet-1k',
    11221: 'imagenet-21k-miil',   # miil subset of fall11
    11821: 'imagenet-12k',        # timm-specific 12k subset of fall11
    21841: 'imagenet-22k',        # as in fall11.tar
    21842: 'imagenet-22k-ms',     # Microsoft (FocalNet) remap: 1k classes first
    21843: 'imagenet-21k-goog',   # Google's full set: two classes not in fall11
}

_SUBSETS = {
    'imagenet1k': 'imagenet_synsets.txt',
    'imagenet12k': 'imagenet12k_synsets.txt',
    'imagenet22k': 'imagenet22k_synsets.txt',
    'imagenet21k': 'imagenet21k_goog_synsets.txt',
    'imagenet21kgoog': 'imagenet21k_goog_synsets.txt',
    'imagenet21kmiil': 'imagenet21k_miil_synsets.txt',
    'imagenet22kms': 'imagenet22k_ms_synsets.txt',
}
_LEMMA_FILE = 'imagenet_synset_to_lemma.txt'
_DEFINITION_FILE = 'imagenet_synset_to_definition.txt'


def infer_imagenet_subset(model_or_cfg) -> Optional[str]:
    """Guess the ImageNet subset from a model / cfg's num_classes."""
    if isinstance(model_or_cfg, dict):
        num_classes = model_or_cfg.get('num_classes', None)
    else:
        num_classes = getattr(model_or_cfg, 'num_classes', None)
        if not num_classes:
            pretrained_cfg = getattr(model_or_cfg, 'pretrained_cfg', {})
            num_classes = pretrained_cfg.get('num_classes', None)
    if not num_classes or num_classes not in _NUM_CLASSES_TO_SUBSET:
        return None
    return _NUM_CLASSES_TO_SUBSET[num_classes]


def _load_tsv(name: str) -> Dict[str, str]:
    data = pkgutil.get_data(__name__, os.path.join('_info', name))
    return dict(csv.reader(data.decode('utf-8').splitlines(), delimiter='\t'))


class ImageNetInfo(DatasetInfo):

    def __init__(self, subset: str = 'imagenet-1k'):
        super().__init__()
        subset = re.sub(r'[-_\s]', '', subset.lower())
        assert subset in _SUBSETS, f'Unknown imagenet subset {subset}.'

        # synsets (pos + wordnet offset) are the canonical class names
        synset_data = pkgutil.get_data(__name__, os.path.join('_info', _SUBSETS[subset]))
        self._synsets = synset_data.decode('utf-8').splitlines()

        # lemmas give the short description, definitions the detailed one
        self._lemmas = _load_tsv(_LEMMA_FILE)
        self._definitions = _load_tsv(_DEFINITION_FILE)

    def num_classes(self):
        return len(self._synsets)

    def label_names(self):
        return self._synsets

    def label_descriptions(self, detailed: bool = False, as_dict: bool = False) -> Union[List[str], Dict[str, str]]:
        if as_dict:
            return {label: self.label_name_to_description(label, detailed=detailed) for label in self._synsets}
        return [self.label_name_to_description(label, detailed=detailed) for label in self._synsets]

    def index_to_label_name(self, index) -> str:
        assert 0 <= index < len(self._synsets), \
            f'Index ({index}) out of range for dataset with {len(self._synsets)} classes.'
        return self._synsets[index]

    def index_to_description(self, index: int, detailed: bool = False) -> str:
        return self.label_name_to_description(self.index_to_label_name(index), detailed=detailed)

    def label_name_to_description(self, label: str, detailed: bool = False) -> str:
        if detailed:
            return f'{self._lemmas[label]}: {self._definitions[label]}'
        return f'{self._lemmas[label]}'
