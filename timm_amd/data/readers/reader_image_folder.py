"""Folder-tree image reader: class labels from directory names.

Behavioral parity: /root/reference/timm/data/readers/reader_image_folder.py
(same walk order, natural-key sorting, class-index assignment).
"""
import os
from typing import Dict, List, Optional, Set, Tuple, Union

from timm_amd.utils.misc import natural_key

from .class_map import load_class_map
from .img_extensions import get_img_extensions
from .reader import Reader


def find_images_and_targets(
        folder: str,
        types: Optional[Union[List, Tuple, Set]] = None,
        class_to_idx: Optional[Dict] = None,
        leaf_name_only: bool = True,
        sort: bool = True,
):
    """Recursively collect (path, class-index) pairs under ``folder``.

    Class names come from the containing directory (leaf name, or the full
    relative path with separators replaced).  Unknown classes (not in a given
    class_to_idx) are dropped.  Returns (samples, class_to_idx).
    """
    exts = set(types) if types else get_img_extensions(as_set=True)
    found: List[Tuple[str, str]] = []  # (path, label)
    for dirpath, _dirs, files in os.walk(folder, topdown=False, followlinks=True):
        if dirpath == folder:
            label = ''
        else:
            rel = os.path.relpath(dirpath, folder)
            label = os.path.basename(rel) if leaf_name_only else rel.replace(os.path.sep, '_')
        found.extend(
            (os.path.join(dirpath, f), label)
            for f in files if os.path.splitext(f)[1].lower() in exts
        )
    if class_to_idx is None:
        names = sorted({label for _, label in found}, key=natural_key)
        class_to_idx = {name: idx for idx, name in enumerate(names)}
    samples = [
        (path, class_to_idx[label]) for path, label in found if label in class_to_idx]
    if sort:
        samples.sort(key=lambda s: natural_key(s[0]))
    return samples, class_to_idx


class ReaderImageFolder(Reader):
    def __init__(self, root, class_map='', input_key=None):
        super().__init__()
        self.root = root
        self.samples, self.class_to_idx = find_images_and_targets(
            root,
            class_to_idx=load_class_map(class_map, root) if class_map else None,
            types=input_key.split(';') if input_key else None,
        )
        if not self.samples:
            raise RuntimeError(
                f'Found 0 images in subfolders of {root}. '
                f'Supported image extensions are {", ".join(get_img_extensions())}')

    def __getitem__(self, index):
        path, target = self.samples[index]
        return open(path, 'rb'), target

    def __len__(self):
        return len(self.samples)

    def _filename(self, index, basename=False, absolute=False):
        path = self.samples[index][0]
        if basename:
            return os.path.basename(path)
        if not absolute:
            return os.path.relpath(path, self.root)
        return path
