from .attention_extract import AttentionExtract
from .agc import adaptive_clip_grad
from .checkpoint_saver import CheckpointSaver
from .clip_grad import dispatch_clip_grad
from .cuda import NativeScaler
from .decay_batch import decay_batch_step, check_batch_size_retry
from .distributed import (
    distribute_bn, init_distributed_device, is_distributed_env, is_primary, reduce_tensor,
    world_info_from_env,
)
from .jit import set_jit_fuser, set_jit_legacy
from .log import setup_default_logging, FormatterNoInfo
from .metrics import AverageMeter, accuracy
from .misc import natural_key, add_bool_arg, ParseKwargs
from .model import (
    unwrap_model, get_state_dict, freeze, unfreeze, reparameterize_model,
    ActivationStatsHook, extract_spp_stats,
)
from .model_ema import ModelEma, ModelEmaV2, ModelEmaV3
from .random import random_seed
from .summary import get_outdir, update_summary
