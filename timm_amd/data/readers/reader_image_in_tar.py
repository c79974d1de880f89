"""Tar-archive image reader: flat .tar, folder of .tar, or .tar-of-.tar.

Behavioral parity: /root/reference/timm/data/readers/reader_image_in_tar.py
(two-level tar nesting, folder-derived labels, pickle tarinfo cache with the
same on-disk dict layout, per-worker cached open tar handles).
"""
import logging
import os
import pickle
import tarfile
from glob import glob
from typing import Dict, List, Optional, Set, Tuple, Union

import numpy as np

from timm_amd.utils.misc import natural_key

from .class_map import load_class_map
from .img_extensions import get_img_extensions
from .reader import Reader

_logger = logging.getLogger(__name__)

CACHE_FILENAME_SUFFIX = '_tarinfos.pickle'


class TarState:
    """Lazily-opened tar handle (+ nested child handles), one per worker."""

    def __init__(self, tf: tarfile.TarFile = None, ti: tarfile.TarInfo = None):
        self.tf = tf
        self.ti = ti
        self.children: Dict[str, 'TarState'] = {}

    def reset(self):
        self.tf = None


def _scan_tar(tf: tarfile.TarFile, node: Dict, extensions: Set[str]) -> int:
    """Stream through ``tf`` filling node['samples'] / node['children'].

    Nested .tar members are scanned one level deep (matches the reference's
    two-level support).  Returns the number of image samples found.
    """
    count = 0
    for idx, ti in enumerate(tf):
        if not ti.isfile():
            continue
        stem, ext = os.path.splitext(os.path.basename(ti.path))
        ext = ext.lower()
        if ext == '.tar':
            child = dict(
                name=ti.name, path=os.path.join(node['path'], stem),
                ti=ti, children=[], samples=[])
            with tarfile.open(fileobj=tf.extractfile(ti), mode='r|') as inner:
                count += _scan_tar(inner, child, extensions)
            _logger.debug(
                f'{idx}/?. Extracted child tarinfos from {ti.name}. '
                f'{len(child["samples"])} images.')
            node['children'].append(child)
        elif ext in extensions:
            node['samples'].append(ti)
            count += 1
    return count


def _scan_all(root, tar_filenames, root_is_tar, root_name, cache_tarinfo, extensions):
    """Scan every top-level tar (or load the pickle cache).  Returns the
    info dict {tars: [node, ...]} in the reference's cache layout."""
    tar_bytes = sum(os.path.getsize(f) for f in tar_filenames)
    _logger.info(f'Scanning {tar_bytes/1024**2:.2f}MB of tar files...')

    if cache_tarinfo is None:
        cache_tarinfo = tar_bytes > 10 * 1024 ** 3  # cache once scans get slow (>10GB)
    cache_path = os.path.join(root, '_' + root_name + CACHE_FILENAME_SUFFIX) \
        if cache_tarinfo else ''

    if cache_path and os.path.exists(cache_path):
        _logger.info(f'Reading tar info from cache file {cache_path}.')
        with open(cache_path, 'rb') as f:
            info = pickle.load(f)
        assert len(info['tars']) == len(tar_filenames), \
            "Cached tartree len doesn't match number of tarfiles"
        return info

    info = dict(tars=[], samples=[])
    for idx, fn in enumerate(tar_filenames):
        path = '' if root_is_tar else os.path.splitext(os.path.basename(fn))[0]
        node = dict(name=os.path.relpath(fn, root), path=path, ti=None, children=[], samples=[])
        with tarfile.open(fn, mode='r|') as tf:  # streaming scan
            n = _scan_tar(tf, node, extensions)
        _logger.debug(
            f'{idx}/{len(tar_filenames)}. Extracted tarinfos from {fn}. '
            f'{len(node["children"])} children, {n} samples.')
        info['tars'].append(node)
    if cache_path:
        _logger.info(f'Writing tar info to cache file {cache_path}.')
        with open(cache_path, 'wb') as f:
            pickle.dump(info, f)
    return info


def extract_tarinfos(
        root,
        class_name_to_idx: Optional[Dict] = None,
        cache_tarinfo: Optional[bool] = None,
        extensions: Optional[Union[List, Tuple, Set]] = None,
        sort: bool = True,
):
    extensions = set(extensions) if extensions else get_img_extensions(as_set=True)
    if os.path.isfile(root):
        assert os.path.splitext(root)[-1].lower() == '.tar'
        tar_filenames = [root]
        root, base = os.path.split(root)
        root_name = os.path.splitext(base)[0]
        root_is_tar = True
    else:
        root_name = root.strip(os.path.sep).split(os.path.sep)[-1]
        tar_filenames = glob(os.path.join(root, '*.tar'), recursive=True)
        root_is_tar = False
    assert tar_filenames, f'No .tar files found at specified path ({root}).'

    info = _scan_all(root, tar_filenames, root_is_tar, root_name, cache_tarinfo, extensions)

    # flatten the (up to two-level) tar tree into parallel sample/label lists
    build_class_map = class_name_to_idx is None
    samples, labels, tarfiles = [], [], []

    def _label_of(node_path, member_dir, leaf_only=True):
        full = os.path.join(node_path, member_dir).strip(os.path.sep)
        return full.split(os.path.sep)[-1] if leaf_only else full.replace(os.path.sep, '_')

    def _collect(node, parent_fn):
        added = 0
        for ti in node['samples']:
            label = _label_of(node['path'], os.path.dirname(ti.path))
            if not build_class_map and label not in class_name_to_idx:
                continue
            samples.append((ti, parent_fn, node['ti']))
            labels.append(label)
            added += 1
        return added

    _logger.info('Collecting samples and building tar states.')
    for node in info['tars']:
        parent_fn = None if root_is_tar else node['name']
        state = TarState()
        added = 0
        for child in node['children']:
            child_added = _collect(child, parent_fn)
            if child_added:
                state.children[child['name']] = TarState(ti=child['ti'])
            added += child_added
        added += _collect(node, parent_fn)
        if added:
            tarfiles.append((parent_fn, state))
    del info

    if build_class_map:
        names = sorted(set(labels), key=natural_key)
        class_name_to_idx = {name: idx for idx, name in enumerate(names)}

    _logger.info('Mapping targets and sorting samples.')
    pairs = [
        (s, class_name_to_idx[label])
        for s, label in zip(samples, labels) if label in class_name_to_idx]
    if sort:
        pairs.sort(key=lambda p: natural_key(p[0][0].path))
    samples, targets = zip(*pairs)
    _logger.info(
        f'Finished processing {len(samples)} samples across {len(tarfiles)} tar files.')
    return np.array(samples), np.array(targets), class_name_to_idx, tarfiles


class ReaderImageInTar(Reader):
    """Reads (file-like, target) samples out of tar archives with per-worker
    handle caching (each DataLoader worker opens its own tar handles)."""

    def __init__(self, root, class_map='', cache_tarfiles=True, cache_tarinfo=None):
        super().__init__()
        self.root = root
        self.cache_tarfiles = cache_tarfiles
        self.samples, self.targets, self.class_name_to_idx, tarfiles = extract_tarinfos(
            root,
            class_name_to_idx=load_class_map(class_map, root) if class_map else None,
            cache_tarinfo=cache_tarinfo,
        )
        self.class_idx_to_name = {v: k for k, v in self.class_name_to_idx.items()}
        self.root_is_tar = len(tarfiles) == 1 and tarfiles[0][0] is None
        self.tar_state = tarfiles[0][1] if self.root_is_tar else dict(tarfiles)

    def __len__(self):
        return len(self.samples)

    def _open_parent(self, parent_fn, state):
        tf = state.tf if state is not None else None
        if tf is None:
            path = os.path.join(self.root, parent_fn) if parent_fn else self.root
            tf = tarfile.open(path)
            if state is not None:
                state.tf = tf
        return tf

    def __getitem__(self, index):
        sample_ti, parent_fn, child_ti = self.samples[index]
        target = self.targets[index]
        state = None
        if self.cache_tarfiles:
            state = self.tar_state if self.root_is_tar else self.tar_state[parent_fn]
        tf = self._open_parent(parent_fn, state)
        if child_ti is not None:
            child_state = state.children[child_ti.name] if state is not None else None
            ctf = child_state.tf if child_state is not None else None
            if ctf is None:
                ctf = tarfile.open(fileobj=tf.extractfile(child_ti))
                if child_state is not None:
                    child_state.tf = ctf
            tf = ctf
        return tf.extractfile(sample_ti), target

    def _filename(self, index, basename=False, absolute=False):
        name = self.samples[index][0].name
        return os.path.basename(name) if basename else name
