"""Reader protocol: indexable/iterable sample sources feeding the datasets.

Behavioral parity: /root/reference/timm/data/readers/reader.py.
"""
from abc import abstractmethod


class Reader:
    """Base sample source.  Subclasses yield (file-like, target) pairs and
    resolve sample filenames for bookkeeping/CSV output."""

    @abstractmethod
    def _filename(self, index, basename=False, absolute=False):
        ...

    def filename(self, index, basename=False, absolute=False):
        return self._filename(index, basename=basename, absolute=absolute)

    def filenames(self, basename=False, absolute=False):
        return [
            self._filename(i, basename=basename, absolute=absolute)
            for i in range(len(self))
        ]
