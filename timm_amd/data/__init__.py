from .auto_augment import (
    RandAugment, AutoAugment, rand_augment_ops, auto_augment_policy,
    rand_augment_transform, auto_augment_transform, augment_and_mix_transform, AugMixAugment,
)
from .config import resolve_data_config, resolve_model_data_config
from .constants import *
from .dataset_info import CustomDatasetInfo, DatasetInfo, DatasetInfoLabelMapper
from .imagenet_info import ImageNetInfo, infer_imagenet_subset
from .dataset import ImageDataset, IterableImageDataset, AugMixDataset
from .dataset_factory import create_dataset
from .distributed_sampler import OrderedDistributedSampler, RepeatAugSampler
from .scheduled_sampler import ScheduledBatchSampler, ScheduledTransformDataset
from .loader import create_loader, fast_collate, PrefetchLoader, MultiEpochsDataLoader
from .mixup import Mixup, FastCollateMixup, mixup_target, rand_bbox, rand_bbox_minmax
from .naflex_mixup import NaFlexMixup, mix_batch_variable_size, pairwise_mixup_target
from .naflex_random_erasing import PatchRandomErasing
from .random_erasing import RandomErasing
from .real_labels import RealLabelsImagenet
from .transforms import (
    ToNumpy, ToTensor, MaybeToTensor, MaybePILToTensor, str_to_interp_mode, str_to_pil_interp,
    interp_mode_to_str, RandomResizedCropAndInterpolation, CenterCropOrPad, RandomCropOrPad,
    RandomPad, ResizeKeepRatio, TrimBorder,
)
from .transforms_factory import (
    create_transform, transforms_imagenet_eval, transforms_imagenet_train, transforms_noaug_train,
)
from .naflex_dataset import NaFlexCollator, NaFlexMapDatasetWrapper, calculate_naflex_batch_size
from .naflex_loader import NaFlexPrefetchLoader, create_naflex_loader
from .naflex_transforms import (
    Patchify, RandomResizedCropToSequence, ResizeToSequence, get_image_size_for_seq, patchify_image,
)
