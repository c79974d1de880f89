"""Streaming reader over TFDS image-classification datasets (reference
`readers/reader_tfds.py:72`).

TensorFlow / tensorflow_datasets are imported lazily at construction (they are
not bundled with the ROCm image); everything TF touches is CPU-pinned so the
MI355X stays PyTorch's. Shard assignment follows the reference scheme: TF
InputContext sharding for training when there are enough TFRecord shards,
fine-grained even sub-splits otherwise (and always for validation, where
dropped/duplicated samples skew the reduce across ranks).
"""
import math
import os
from typing import Optional

import torch
import torch.distributed as dist
from PIL import Image

from .class_map import load_class_map
from .reader import Reader
from .shared_count import SharedCount

MAX_TP_SIZE = int(os.environ.get('TFDS_TP_SIZE', 8))
SHUFFLE_SIZE = int(os.environ.get('TFDS_SHUFFLE_SIZE', 8192))
PREFETCH_SIZE = int(os.environ.get('TFDS_PREFETCH_SIZE', 2048))

tf = None
tfds = None


def _import_tf():
    global tf, tfds
    if tfds is not None:
        return
    try:
        import tensorflow as _tf
        _tf.config.set_visible_devices([], 'GPU')  # keep TF off the MI355X
        import tensorflow_datasets as _tfds
    except ImportError as e:
        raise RuntimeError(
            'tfds/ datasets need the tensorflow_datasets package '
            '(`pip install tensorflow-datasets tensorflow-cpu`): ' + str(e)) from e
    tf, tfds = _tf, _tfds


def even_split_indices(split, n, num_samples):
    partitions = [round(i * num_samples / n) for i in range(n + 1)]
    return [f'{split}[{partitions[i]}:{partitions[i + 1]}]' for i in range(n)]


def get_class_labels(info):
    if 'label' not in info.features:
        return {}
    class_label = info.features['label']
    return {n: class_label.str2int(n) for n in class_label.names}


class ReaderTfds(Reader):
    """TFDS -> PyTorch iterable bridge. Decompressed single samples come out;
    augmentation and batching stay in the torch pipeline."""

    def __init__(
            self,
            name,
            root=None,
            split='train',
            class_map=None,
            is_training=False,
            batch_size=1,
            download=False,
            repeats=0,
            seed=42,
            input_key='image',
            input_img_mode='RGB',
            target_key='label',
            target_img_mode='',
            prefetch_size=None,
            shuffle_size=None,
            max_threadpool_size=None,
    ):
        super().__init__()
        _import_tf()
        self.root = root
        self.split = split
        self.is_training = is_training
        self.batch_size = batch_size
        self.repeats = repeats
        self.common_seed = seed

        self.prefetch_size = prefetch_size or PREFETCH_SIZE
        self.shuffle_size = shuffle_size or SHUFFLE_SIZE
        self.max_threadpool_size = max_threadpool_size or MAX_TP_SIZE

        self.input_key = input_key
        self.input_img_mode = input_img_mode
        self.target_key = target_key
        self.target_img_mode = target_img_mode
        self.builder = tfds.builder(name, data_dir=root)
        if download:
            self.builder.download_and_prepare()
        self.remap_class = False
        if class_map:
            self.class_to_idx = load_class_map(class_map)
            self.remap_class = True
        else:
            self.class_to_idx = get_class_labels(self.builder.info) if self.target_key == 'label' else {}
        self.split_info = self.builder.info.splits[split]
        self.num_samples = self.split_info.num_examples

        self.dist_rank = 0
        self.dist_num_replicas = 1
        if dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1:
            self.dist_rank = dist.get_rank()
            self.dist_num_replicas = dist.get_world_size()

        self.global_num_workers = 1
        self.num_workers = 1
        self.worker_info = None
        self.worker_seed = 0
        self.subsplit = None
        self.ds = None
        self.init_count = 0
        self.epoch_count = SharedCount()
        # reshuffle semantics across workers/epochs are only trustworthy if the
        # TF pipeline is rebuilt per iteration in training
        self.reinit_each_iter = self.is_training

    def set_epoch(self, count):
        self.epoch_count.value = count

    def set_loader_cfg(self, num_workers: Optional[int] = None):
        if self.ds is not None:
            return
        if num_workers is not None:
            self.num_workers = num_workers
            self.global_num_workers = self.dist_num_replicas * self.num_workers

    def _lazy_init(self):
        """Build the tf.data pipeline inside the DataLoader worker process."""
        worker_info = torch.utils.data.get_worker_info()

        global_worker_id = 0
        if worker_info is not None:
            self.worker_info = worker_info
            self.worker_seed = worker_info.seed
            self.num_workers = worker_info.num_workers
            self.global_num_workers = self.dist_num_replicas * self.num_workers
            global_worker_id = self.dist_rank * self.num_workers + worker_info.id

            # Not enough TFRecord shards for every worker (or validating, where
            # sample counts must stay even) -> fine-grained sub-splits; else TF
            # InputContext sharding assigns whole shards per pipeline.
            should_subsplit = self.global_num_workers > 1 and (
                    self.split_info.num_shards < self.global_num_workers or not self.is_training)
            if should_subsplit:
                subsplits = tfds.even_splits(self.split, self.global_num_workers)
                self.subsplit = subsplits[global_worker_id]

        input_context = None
        if self.global_num_workers > 1 and self.subsplit is None:
            input_context = tf.distribute.InputContext(
                num_input_pipelines=self.global_num_workers,
                input_pipeline_id=global_worker_id,
                num_replicas_in_sync=self.dist_num_replicas,
            )
        read_config = tfds.ReadConfig(
            shuffle_seed=self.common_seed + self.epoch_count.value,
            shuffle_reshuffle_each_iteration=True,
            input_context=input_context,
        )

        @tfds.decode.make_decoder()
        def decode_example(serialized_image, feature, dct_method='INTEGER_ACCURATE', channels=3):
            return tf.image.decode_jpeg(serialized_image, channels=channels, dct_method=dct_method)

        ds = self.builder.as_dataset(
            split=self.subsplit or self.split,
            shuffle_files=self.is_training,
            decoders=dict(image=decode_example(channels=1 if self.input_img_mode == 'L' else 3)),
            read_config=read_config,
        )
        # don't stack TF's threadpools on top of the torch workers
        options = tf.data.Options()
        thread_member = 'threading' if hasattr(options, 'threading') else 'experimental_threading'
        getattr(options, thread_member).private_threadpool_size = max(1, self.max_threadpool_size // self.num_workers)
        getattr(options, thread_member).max_intra_op_parallelism = 1
        ds = ds.with_options(options)
        if self.is_training or self.repeats > 1:
            ds = ds.repeat()  # wrap around; iteration is cut at the sample budget
        if self.is_training:
            ds = ds.shuffle(
                min(self.num_samples, self.shuffle_size) // self.global_num_workers, seed=self.worker_seed)
        ds = ds.prefetch(min(self.num_samples // self.global_num_workers, self.prefetch_size))
        self.ds = tfds.as_numpy(ds)
        self.init_count += 1

    def _num_samples_per_worker(self):
        num_worker_samples = (
            max(1, self.repeats) * self.num_samples / max(self.global_num_workers, self.dist_num_replicas))
        if self.is_training or self.dist_num_replicas > 1:
            num_worker_samples = math.ceil(num_worker_samples)
        if self.is_training:
            num_worker_samples = math.ceil(num_worker_samples / self.batch_size) * self.batch_size
        return int(num_worker_samples)

    def __iter__(self):
        if self.ds is None or self.reinit_each_iter:
            self._lazy_init()

        target_sample_count = self._num_samples_per_worker()
        sample_count = 0
        input_data = target_data = None
        for sample in self.ds:
            input_data = sample[self.input_key]
            if self.input_img_mode:
                if self.input_img_mode == 'L' and input_data.ndim == 3:
                    input_data = input_data[:, :, 0]
                input_data = Image.fromarray(input_data, mode=self.input_img_mode)
            target_data = sample[self.target_key]
            if self.target_img_mode:
                target_data = Image.fromarray(target_data, mode=self.target_img_mode)
            elif self.remap_class:
                target_data = self.class_to_idx[target_data]
            yield input_data, target_data
            sample_count += 1
            if self.is_training and sample_count >= target_sample_count:
                break

        # distributed validation: pad ranks to equal counts by replaying the
        # last sample (mirrors the reference's sub-split padding caveats)
        if not self.is_training and self.dist_num_replicas > 1 and self.subsplit is not None and \
                0 < sample_count < target_sample_count:
            while sample_count < target_sample_count:
                yield input_data, target_data
                sample_count += 1

    def __len__(self):
        return self._num_samples_per_worker() * self.num_workers

    def _filename(self, index, basename=False, absolute=False):
        raise AssertionError('Not supported')  # no random access

    def filenames(self, basename=False, absolute=False):
        if self.ds is None:
            self._lazy_init()
        names = []
        for sample in self.ds:
            if len(names) > self.num_samples:
                break
            for key in ('file_name', 'filename', 'id'):
                if key in sample:
                    names.append(sample[key])
                    break
            else:
                raise AssertionError('No supported name field present')
        return names
