"""Map-style reader over Hugging Face `datasets` (reference
`readers/reader_hfds.py:30`). Decode is left to the transform pipeline — the
column is re-cast with decode=False so samples come back as raw bytes/paths.
"""
import io
from typing import Optional

try:
    import datasets
except ImportError as e:
    datasets = None
    _DATASETS_ERR = e

from .class_map import load_class_map
from .reader import Reader


def get_class_labels(info, label_key='label'):
    if 'label' not in info.features:
        return {}
    class_label = info.features[label_key]
    return {n: class_label.str2int(n) for n in class_label.names}


class ReaderHfds(Reader):
    def __init__(
            self,
            name: str,
            root: Optional[str] = None,
            split: str = 'train',
            class_map: dict = None,
            input_key: str = 'image',
            target_key: str = 'label',
            additional_features: Optional[list] = None,
            download: bool = False,
            trust_remote_code: bool = False,
    ):
        super().__init__()
        if datasets is None:
            raise RuntimeError(
                f'Hugging Face datasets package is required for hfds/ datasets: {_DATASETS_ERR}')
        self.root = root
        self.split = split
        self.dataset = datasets.load_dataset(
            name,
            split=split,
            cache_dir=self.root,
            trust_remote_code=trust_remote_code,
        )
        self.dataset = self.dataset.cast_column(input_key, datasets.Image(decode=False))

        self.image_key = input_key
        self.label_key = target_key
        self.remap_class = False
        if class_map:
            self.class_to_idx = load_class_map(class_map)
            self.remap_class = True
        else:
            self.class_to_idx = get_class_labels(self.dataset.info, self.label_key)
        self.split_info = self.dataset.info.splits[split]
        self.num_samples = self.split_info.num_examples
        if additional_features is not None:
            self.additional_features = (
                additional_features if isinstance(additional_features, list) else [additional_features])
        else:
            self.additional_features = None

    def __getitem__(self, index):
        item = self.dataset[index]
        image = item[self.image_key]
        if 'bytes' in image and image['bytes']:
            image = io.BytesIO(image['bytes'])
        else:
            assert 'path' in image and image['path']
            image = open(image['path'], 'rb')
        label = item[self.label_key]
        if self.remap_class:
            label = self.class_to_idx[label]
        if self.additional_features is not None:
            return (image, label, *[item[feat] for feat in self.additional_features])
        return image, label

    def __len__(self):
        return len(self.dataset)

    def _filename(self, index, basename=False, absolute=False):
        return self.dataset[index][self.image_key]['path']
