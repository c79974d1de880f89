"""Webdataset-format shard reader (reference `readers/reader_wds.py:262`).

Reads datasets stored as directories of `.tar` shards where each sample is a
group of files sharing a basename key (`00001.jpg`, `00001.cls`, ...), with an
`_info.json` describing splits. Unlike the reference this is self-contained —
shard iteration, brace expansion, grouping and the deterministic shuffles are
implemented directly on `tarfile` instead of depending on the `webdataset`
package (one less moving part, and the pipeline stays a flat generator chain
feeding the torch DataLoader).
"""
import io
import json
import logging
import math
import os
import random
import re
import tarfile
from dataclasses import dataclass
from itertools import islice
from typing import Dict, Iterator, Optional, Tuple

import torch
import torch.distributed as dist
import yaml
from PIL import Image

from .class_map import load_class_map
from .reader import Reader
from .shared_count import SharedCount

_logger = logging.getLogger(__name__)

SAMPLE_SHUFFLE_SIZE = int(os.environ.get('WDS_SHUFFLE_SIZE', 8192))
SAMPLE_INITIAL_SIZE = int(os.environ.get('WDS_INITIAL_SIZE', 2048))

_BRACE_RE = re.compile(r'\{(\d+)\.\.(\d+)\}')


def expand_urls(spec: str):
    """Expand `prefix-{0000..0099}.tar`-style brace ranges (zero-padded)."""
    m = _BRACE_RE.search(spec)
    if not m:
        return [spec]
    lo, hi = m.group(1), m.group(2)
    width = len(lo)
    out = []
    for i in range(int(lo), int(hi) + 1):
        out.extend(expand_urls(spec[:m.start()] + f'{i:0{width}d}' + spec[m.end():]))
    return out


def _load_info(root, names=('_info.json', 'info.json', '_info.yaml', 'info.yaml')):
    if isinstance(names, str):
        names = (names,)
    tried = []
    err_str = ''
    for n in names:
        full_path = os.path.join(root, n)
        tried.append(full_path)
        try:
            with open(full_path) as f:
                if n.endswith('.json'):
                    return json.load(f)
                return yaml.safe_load(f)
        except Exception as e:
            err_str = str(e)
    _logger.warning(
        f'Dataset info file not found at {tried}. Error: {err_str}. '
        'Falling back to provided split and size arg.')
    return {}


@dataclass
class SplitInfo:
    num_samples: int
    filenames: Tuple[str]
    shard_lengths: Tuple[int] = ()
    alt_label: str = ''
    name: str = ''


def _parse_split_info(split: str, info: Dict) -> SplitInfo:
    def _info_convert(dict_info):
        return SplitInfo(
            num_samples=dict_info['num_samples'],
            filenames=tuple(dict_info['filenames']),
            shard_lengths=tuple(dict_info['shard_lengths']),
            alt_label=dict_info.get('alt_label', ''),
            name=dict_info['name'],
        )

    if 'tar' in split or '..' in split:
        # brace-expand form, optional `|num_samples` suffix:
        # `dataset-train-{0000..9999}.tar|100000`
        split = split.split('|')
        num_samples = 0
        split_name = ''
        if len(split) > 1:
            num_samples = int(split[1])
        split = split[0]
        if '::' not in split:
            split_parts = split.split('-', 3)
            split_idx = len(split_parts) - 1
            if split_idx and 'splits' in info and split_parts[split_idx] in info['splits']:
                split_name = split_parts[split_idx]

        split_filenames = expand_urls(split)
        if split_name:
            split_info = info['splits'][split_name]
            if not num_samples:
                _fc = {f: c for f, c in zip(split_info['filenames'], split_info['shard_lengths'])}
                num_samples = sum(_fc[f] for f in split_filenames)
                split_info['filenames'] = tuple(_fc.keys())
                split_info['shard_lengths'] = tuple(_fc.values())
                split_info['num_samples'] = num_samples
            split_info = _info_convert(split_info)
        else:
            split_info = SplitInfo(
                name=split_name,
                num_samples=num_samples,
                filenames=tuple(split_filenames),
            )
    else:
        if 'splits' not in info or split not in info['splits']:
            raise RuntimeError(f"split {split} not found in info ({info.get('splits', {}).keys()})")
        split_info = _info_convert(info['splits'][split])
    return split_info


def _tar_samples(path: str) -> Iterator[Dict[str, bytes]]:
    """Stream one shard: group consecutive members by their sample key
    (basename up to the first dot) into {ext: bytes} dicts."""
    current_key = None
    sample = {}
    try:
        with tarfile.open(path, mode='r|*') as tf:  # streaming mode: one sequential pass
            for ti in tf:
                if not ti.isfile():
                    continue
                basename = os.path.basename(ti.name)
                if basename.startswith('.'):
                    continue
                key, _, ext = basename.partition('.')
                if key != current_key:
                    if current_key is not None and sample:
                        sample['__key__'] = current_key
                        yield sample
                    current_key = key
                    sample = {}
                data = tf.extractfile(ti)
                if data is not None:
                    sample[ext.lower()] = data.read()
    except Exception as exn:
        _logger.warning(f'Handling webdataset shard error ({repr(exn)}) in {path}. Ignoring.')
        if isinstance(exn, TypeError):
            raise
    if current_key is not None and sample:
        sample['__key__'] = current_key
        yield sample


def _buffered_shuffle(src, bufsize: int, initial: int, rng: random.Random):
    """Streaming reservoir-style shuffle (same contract as wds.filters._shuffle)."""
    buf = []
    for x in src:
        if len(buf) < bufsize:
            buf.append(x)
            if len(buf) < initial:
                continue
        idx = rng.randint(0, len(buf) - 1)
        buf[idx], x = x, buf[idx]
        yield x
    rng.shuffle(buf)
    yield from buf


def _getfirst(sample: Dict, keys: str):
    """First present key from a ';'-separated preference list."""
    for k in keys.split(';'):
        if k in sample:
            return sample[k]
    return None


def _decode(sample, image_key='jpg', image_mode='RGB', target_key='cls', alt_label=''):
    """Decode one raw sample: PIL image, int label, JSON passed through raw."""
    if alt_label:
        meta = json.loads(sample['json'])
        class_label = int(meta[alt_label])
        if class_label < 0:
            return None  # skipped labels encoded as -1
    else:
        raw = _getfirst(sample, target_key)
        if raw is None:
            return None
        class_label = int(raw)

    img = _getfirst(sample, image_key)
    if img is None:
        return None
    with io.BytesIO(img) as b:
        img = Image.open(b)
        img.load()
    if image_mode:
        img = img.convert(image_mode)
    return dict(image=img, target=class_label, json=sample.get('json', None), __key__=sample['__key__'])


class ReaderWds(Reader):
    def __init__(
            self,
            root: str,
            name: Optional[str] = None,
            split: str = 'train',
            is_training: bool = False,
            num_samples: Optional[int] = None,
            batch_size: int = 1,
            repeats: int = 0,
            seed: int = 42,
            class_map: Optional[dict] = None,
            input_key: str = 'jpg;png;webp',
            input_img_mode: str = 'RGB',
            target_key: str = 'cls',
            target_img_mode: str = '',
            filename_key: str = 'filename',
            sample_shuffle_size: Optional[int] = None,
            sample_initial_size: Optional[int] = None,
    ):
        super().__init__()
        self.root = root
        self.is_training = is_training
        self.batch_size = batch_size
        self.repeats = repeats
        self.common_seed = seed  # fixed across all workers/ranks
        self.shard_shuffle_size = 500
        self.sample_shuffle_size = sample_shuffle_size or SAMPLE_SHUFFLE_SIZE
        self.sample_initial_size = sample_initial_size or SAMPLE_INITIAL_SIZE

        self.input_key = input_key
        self.input_img_mode = input_img_mode
        self.target_key = target_key
        self.filename_key = filename_key
        self.key_ext = '.JPEG'  # added to __key__ for original filenames (ImageNet default)

        self.info = _load_info(self.root)
        self.split_info = _parse_split_info(split, self.info)
        self.num_samples = num_samples if num_samples is not None else self.split_info.num_samples
        if is_training and not self.num_samples:
            raise RuntimeError('Invalid split definition, num_samples not specified in train mode.')

        self.remap_class = False
        if class_map:
            self.class_to_idx = load_class_map(class_map)
            self.remap_class = True
        else:
            self.class_to_idx = {}

        self.dist_rank = 0
        self.dist_num_replicas = 1
        if dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1:
            self.dist_rank = dist.get_rank()
            self.dist_num_replicas = dist.get_world_size()

        # filled in by _lazy_init on each DataLoader worker
        self.worker_info = None
        self.worker_id = 0
        self.worker_seed = seed
        self.num_workers = 1
        self.global_worker_id = 0
        self.global_num_workers = 1
        self.epoch_count = SharedCount()
        self._initialized = False

    def set_epoch(self, count):
        self.epoch_count.value = count

    def set_loader_cfg(self, num_workers: Optional[int] = None):
        if self._initialized:
            return
        if num_workers is not None:
            self.num_workers = num_workers
            self.global_num_workers = self.dist_num_replicas * self.num_workers

    def _lazy_init(self):
        if self.worker_info is None:
            worker_info = torch.utils.data.get_worker_info()
            if worker_info is not None:
                self.worker_info = worker_info
                self.worker_id = worker_info.id
                self.worker_seed = worker_info.seed
                self.num_workers = worker_info.num_workers
            self.global_num_workers = self.dist_num_replicas * self.num_workers
            self.global_worker_id = self.dist_rank * self.num_workers + self.worker_id
        self._initialized = True

    def _shard_paths(self, epoch: int):
        """Shard list for this epoch: epoch-seeded shuffle (training), then
        this worker's strided slice of the global list."""
        paths = [os.path.join(self.root, f) for f in self.split_info.filenames]
        if self.is_training:
            rng = random.Random(self.common_seed + epoch)
            paths = list(paths)
            rng.shuffle(paths)
        if self.global_num_workers > 1:
            paths = list(islice(paths, self.global_worker_id, None, self.global_num_workers))
        return paths

    def _samples(self, epoch: int):
        """Decoded sample stream for this worker: shards -> grouped samples
        -> (training) buffered shuffle -> decode."""
        def raw():
            shards = self._shard_paths(epoch)
            while True:
                for path in shards:
                    yield from _tar_samples(path)
                if not self.is_training:
                    return
                # training wraps across the shard list until the budget is met

        src = raw()
        if self.is_training:
            src = _buffered_shuffle(
                src, self.sample_shuffle_size, self.sample_initial_size, random.Random(self.worker_seed))
        for sample in src:
            decoded = _decode(
                sample,
                image_key=self.input_key,
                image_mode=self.input_img_mode,
                target_key=self.target_key,
                alt_label=self.split_info.alt_label,
            )
            if decoded is not None:
                yield decoded

    def _num_samples_per_worker(self):
        num_worker_samples = self.num_samples / max(self.global_num_workers, self.dist_num_replicas)
        if self.is_training or self.dist_num_replicas > 1:
            num_worker_samples = math.ceil(num_worker_samples)
        if self.is_training:
            num_worker_samples = math.ceil(num_worker_samples / self.batch_size) * self.batch_size
        return int(num_worker_samples)

    def __iter__(self):
        if not self._initialized:
            self._lazy_init()
        budget = self._num_samples_per_worker()
        limited = self.is_training or self.dist_num_replicas > 1
        count = 0
        for sample in self._samples(self.epoch_count.value):
            target = sample['target']
            if self.remap_class:
                target = self.class_to_idx[target]
            yield sample['image'], target
            count += 1
            if limited and count >= budget:
                break

    def __len__(self):
        return self._num_samples_per_worker() * self.num_workers

    def _filename(self, index, basename=False, absolute=False):
        raise AssertionError('Not supported')  # no random access

    def filenames(self, basename=False, absolute=False):
        if not self._initialized:
            self._lazy_init()
        names = []
        for sample in self._samples(self.epoch_count.value):
            if self.filename_key in sample:
                names.append(sample[self.filename_key])
            else:
                names.append(sample['__key__'] + self.key_ext)
            if len(names) >= self.num_samples:
                break
        return names
