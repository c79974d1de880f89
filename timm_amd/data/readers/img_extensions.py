"""Mutable registry of recognised image-file extensions (reference
`timm/data/readers/img_extensions.py`). Folder/tar readers consult it when
scanning for samples."""
from copy import deepcopy

__all__ = [
    'get_img_extensions', 'is_img_extension', 'set_img_extensions',
    'add_img_extensions', 'del_img_extensions',
]

IMG_EXTENSIONS = ('.png', '.jpg', '.jpeg')  # ordered public tuple (bwd compat)
_IMG_EXTENSIONS_SET = set(IMG_EXTENSIONS)   # membership-test twin, kept in sync


def _rebuild(extensions):
    global IMG_EXTENSIONS, _IMG_EXTENSIONS_SET
    seen = set()
    ordered = []
    for ext in extensions:
        if ext not in seen:
            seen.add(ext)
            ordered.append(ext)
    IMG_EXTENSIONS = tuple(ordered)
    _IMG_EXTENSIONS_SET = seen


def _check(ext) -> bool:
    return bool(ext) and isinstance(ext, str) and len(ext) >= 2 and ext[0] == '.'


def is_img_extension(ext):
    return ext in _IMG_EXTENSIONS_SET


def get_img_extensions(as_set=False):
    return deepcopy(_IMG_EXTENSIONS_SET if as_set else IMG_EXTENSIONS)


def set_img_extensions(extensions):
    assert len(extensions)
    assert all(_check(x) for x in extensions)
    _rebuild(extensions)


def add_img_extensions(ext):
    if not isinstance(ext, (list, tuple, set)):
        ext = (ext,)
    assert all(_check(x) for x in ext)
    _rebuild(tuple(IMG_EXTENSIONS) + tuple(ext))


def del_img_extensions(ext):
    if not isinstance(ext, (list, tuple, set)):
        ext = (ext,)
    remove = set(ext)
    _rebuild(x for x in IMG_EXTENSIONS if x not in remove)
