"""timm_amd — MI355X-native image-model training & inference framework.

Brand-new framework with the capabilities of huggingface/pytorch-image-models
(timm v1.0.28 reference at /root/reference), rebuilt MI355X-first:
PyTorch-ROCm model graphs + hand-written gfx950 HIP kernels for the hot path
+ RCCL-over-xGMI data parallelism.
"""
from .version import __version__
from .layers import (
    is_scriptable, is_exportable, set_scriptable, set_exportable,
)
from .models import (
    create_model, list_models, list_pretrained, is_model, list_modules, model_entrypoint,
    is_model_pretrained, get_pretrained_cfg, get_pretrained_cfg_value,
)
