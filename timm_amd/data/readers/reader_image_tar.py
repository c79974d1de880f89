"""Single-tarfile dataset reader (reference `readers/reader_image_tar.py`).

Images live inside one `.tar`; the parent folder of each member is its class.
Superseded by `ReaderImageInTar` for folder-of-tars layouts, kept for parity.
"""
import os
import tarfile

from ...utils.misc import natural_key
from .class_map import load_class_map
from .img_extensions import get_img_extensions
from .reader import Reader


def extract_tarinfo(tf: tarfile.TarFile, class_to_idx=None, sort: bool = True):
    extensions = get_img_extensions(as_set=True)
    files = []
    labels = []
    for ti in tf.getmembers():
        if not ti.isfile():
            continue
        dirname, basename = os.path.split(ti.path)
        label = os.path.basename(dirname)
        ext = os.path.splitext(basename)[1]
        if ext.lower() in extensions:
            files.append(ti)
            labels.append(label)
    if class_to_idx is None:
        sorted_labels = sorted(set(labels), key=natural_key)
        class_to_idx = {c: idx for idx, c in enumerate(sorted_labels)}
    samples = [(f, class_to_idx[l]) for f, l in zip(files, labels) if l in class_to_idx]
    if sort:
        samples = sorted(samples, key=lambda k: natural_key(k[0].path))
    return samples, class_to_idx


class ReaderImageTar(Reader):
    def __init__(self, root, class_map=''):
        super().__init__()
        class_to_idx = None
        if class_map:
            class_to_idx = load_class_map(class_map, root)
        assert os.path.isfile(root)
        self.root = root
        with tarfile.open(root) as tf:
            # the handle cannot cross into DataLoader workers; reopen lazily
            self.samples, self.class_to_idx = extract_tarinfo(tf, class_to_idx)
        self.imgs = self.samples
        self.tarfile = None

    def __getitem__(self, index):
        if self.tarfile is None:
            self.tarfile = tarfile.open(self.root)
        tarinfo, target = self.samples[index]
        return self.tarfile.extractfile(tarinfo), target

    def __len__(self):
        return len(self.samples)

    def _filename(self, index, basename=False, absolute=False):
        filename = self.samples[index][0].name
        if basename:
            filename = os.path.basename(filename)
        return filename
