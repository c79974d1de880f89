"""Dataset label metadata interfaces (reference `timm/data/dataset_info.py`).

`DatasetInfo` is the abstract label-name/description provider consumed by the
inference CLI (`--label-type`); `CustomDatasetInfo` wraps user-supplied label
containers (list/tuple, or possibly-sparse index->name dict as loaded from
hub JSON configs); `DatasetInfoLabelMapper` turns classifier indices into
display labels with a fallback for unmapped indices.
"""
from abc import ABC, abstractmethod
from numbers import Integral
from typing import Dict, List, NamedTuple, Optional, Tuple, Union

LabelNames = Union[List[str], Tuple[str, ...], Dict[Union[int, str], str]]


class DatasetInfo(ABC):

    @abstractmethod
    def num_classes(self):
        ...

    @abstractmethod
    def label_names(self):
        ...

    def label_indices(self) -> Tuple[int, ...]:
        """Classifier indices that have label names."""
        return tuple(range(self.num_classes()))

    def has_label(self, index) -> bool:
        return 0 <= int(index) < self.num_classes()

    @abstractmethod
    def label_descriptions(self, detailed: bool = False, as_dict: bool = False) -> Union[List[str], Dict[str, str]]:
        ...

    @abstractmethod
    def index_to_label_name(self, index) -> str:
        ...

    @abstractmethod
    def index_to_description(self, index: int, detailed: bool = False) -> str:
        ...

    @abstractmethod
    def label_name_to_description(self, label: str, detailed: bool = False) -> str:
        ...


class CustomDatasetInfo(DatasetInfo):
    """DatasetInfo over caller-supplied labels; dict label maps may be sparse."""

    def __init__(
            self,
            label_names: LabelNames,
            label_descriptions: Optional[Dict[str, str]] = None,
    ):
        super().__init__()
        if not isinstance(label_names, (list, tuple, dict)) or not label_names:
            raise ValueError('label_names must be a non-empty list, tuple, or index-to-name dictionary.')

        self._label_names = label_names  # original container kept for callers
        self._label_names_by_index = None
        if isinstance(label_names, dict):
            # hub JSON configs deliver string keys; normalize to ints
            by_index = {}
            for raw_index, label_name in label_names.items():
                if not isinstance(raw_index, (Integral, str)):
                    raise TypeError(f'Label index must be an int or string, got {type(raw_index).__name__}.')
                try:
                    index = int(raw_index)
                except ValueError as e:
                    raise ValueError(f'Label index must be integer-like, got {raw_index!r}.') from e
                if index < 0:
                    raise ValueError(f'Label index must be non-negative, got {index}.')
                if index in by_index:
                    raise ValueError(f'Duplicate label index after normalization: {index}.')
                by_index[index] = label_name
            self._label_names_by_index = by_index
            name_values = by_index.values()
        else:
            name_values = label_names

        if not all(isinstance(name, str) for name in name_values):
            raise TypeError('All label names must be strings.')

        self._label_descriptions = label_descriptions
        if self._label_descriptions is not None:
            if not isinstance(self._label_descriptions, dict):
                raise TypeError('label_descriptions must be a label-name-to-description dictionary.')
            missing = [name for name in name_values if name not in self._label_descriptions]
            if missing:
                raise ValueError(f'Missing descriptions for label names: {missing}.')

    def num_classes(self):
        return len(self._label_names)

    def label_names(self):
        return self._label_names

    def label_indices(self) -> Tuple[int, ...]:
        if self._label_names_by_index is not None:
            return tuple(self._label_names_by_index)
        return tuple(range(len(self._label_names)))

    def has_label(self, index) -> bool:
        index = int(index)
        if self._label_names_by_index is not None:
            return index in self._label_names_by_index
        return 0 <= index < len(self._label_names)

    def label_descriptions(self, detailed: bool = False, as_dict: bool = False) -> Union[List[str], Dict[str, str]]:
        return self._label_descriptions

    def label_name_to_description(self, label: str, detailed: bool = False) -> str:
        if self._label_descriptions:
            return self._label_descriptions[label]
        return label  # no description table: the name is its own description

    def index_to_label_name(self, index) -> str:
        if self._label_names_by_index is not None:
            return self._label_names_by_index[int(index)]
        assert 0 <= index < len(self._label_names)
        return self._label_names[index]

    def index_to_description(self, index: int, detailed: bool = False) -> str:
        return self.label_name_to_description(self.index_to_label_name(index), detailed=detailed)


class LabelMappingCoverage(NamedTuple):
    mapped: int
    missing: int
    extra: int


class DatasetInfoLabelMapper:
    """index -> display-label callable with sparse-mapping fallback."""

    def __init__(
            self,
            dataset_info: DatasetInfo,
            label_type: str = 'description',
            fallback_format: Optional[str] = '<unmapped:{index}>',
    ):
        if label_type not in ('name', 'description', 'detail', 'detailed'):
            raise ValueError(f'Invalid label type: {label_type}.')
        self.dataset_info = dataset_info
        self.label_type = label_type
        self.fallback_format = fallback_format

    def __call__(self, index) -> str:
        index = int(index)
        if not self.dataset_info.has_label(index) and self.fallback_format is not None:
            return self.fallback_format.format(index=index)
        if self.label_type == 'name':
            return self.dataset_info.index_to_label_name(index)
        return self.dataset_info.index_to_description(
            index, detailed=self.label_type in ('detail', 'detailed'))

    def coverage(self, num_classes: int) -> LabelMappingCoverage:
        """(mapped, missing, extra) counts for a classifier of num_classes."""
        if num_classes < 0:
            raise ValueError(f'num_classes must be non-negative, got {num_classes}.')
        mapped_indices = self.dataset_info.label_indices()
        mapped = sum(0 <= index < num_classes for index in mapped_indices)
        return LabelMappingCoverage(
            mapped=mapped,
            missing=num_classes - mapped,
            extra=len(mapped_indices) - mapped,
        )
