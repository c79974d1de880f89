"""Reader factory: '<prefix>/<name>' spec -> reader instance.

Behavioral parity: /root/reference/timm/data/readers/reader_factory.py:8.
Streaming backends import lazily (their deps are optional).
"""
import os
from typing import Optional


def _hfds(name, root, split, **kwargs):
    from .reader_hfds import ReaderHfds
    return ReaderHfds(name=name, root=root, split=split, **kwargs)


def _hfids(name, root, split, **kwargs):
    from .reader_hfids import ReaderHfids
    return ReaderHfids(name=name, root=root, split=split, **kwargs)


def _tfds(name, root, split, **kwargs):
    from .reader_tfds import ReaderTfds
    return ReaderTfds(name=name, root=root, split=split, **kwargs)


def _wds(name, root, split, **kwargs):
    from .reader_wds import ReaderWds
    kwargs.pop('download', False)
    return ReaderWds(root=root, name=name, split=split, **kwargs)


_STREAMING = dict(hfds=_hfds, hfids=_hfids, tfds=_tfds, wds=_wds)


def create_reader(name: str, root: Optional[str] = None, split: str = 'train', **kwargs):
    kwargs = {k: v for k, v in kwargs.items() if v is not None}
    prefix, sep, base = name.lower().partition('/')
    if not sep:
        prefix, base = '', prefix

    if prefix in _STREAMING:
        return _STREAMING[prefix](base, root, split, **kwargs)

    # local-path fallback: a .tar file or a directory tree of images
    assert os.path.exists(root)
    if os.path.isfile(root) and os.path.splitext(root)[1] == '.tar':
        from .reader_image_in_tar import ReaderImageInTar
        return ReaderImageInTar(root, **kwargs)
    from .reader_image_folder import ReaderImageFolder
    return ReaderImageFolder(root, **kwargs)
