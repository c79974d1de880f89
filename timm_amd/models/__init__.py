from ._builder import (
    build_model_with_cfg, load_custom_pretrained, load_pretrained, pretrained_cfg_for_features,
    resolve_pretrained_cfg, set_pretrained_download_progress, set_pretrained_check_hash,
)
from ._factory import create_model, parse_model_name, safe_model_name
from ._features import (
    FeatureInfo, FeatureHooks, FeatureDictNet, FeatureListNet, FeatureHookNet, FeatureGetterNet,
    feature_take_indices,
)
from ._features_fx import (
    FeatureGraphNet, GraphExtractNet, create_feature_extractor, get_graph_node_names,
    register_notrace_module, is_notrace_module, get_notrace_modules,
    register_notrace_function, is_notrace_function, get_notrace_functions,
)
from ._helpers import clean_state_dict, load_state_dict, load_checkpoint, remap_state_dict, resume_checkpoint
from ._hub import (
    load_model_config_from_hf, load_state_dict_from_hf, push_to_hf_hub,
)
from ._manipulate import (
    model_parameters, named_apply, named_modules, named_modules_with_params, group_modules,
    group_parameters, checkpoint, checkpoint_seq, adapt_input_conv,
)
from ._pretrained import PretrainedCfg, DefaultCfg, filter_pretrained_cfg
from ._prune import adapt_model_from_string
from ._registry import (
    split_model_name_tag, get_arch_name, generate_default_cfgs, register_model,
    register_model_deprecations, model_entrypoint, list_models, list_pretrained, get_deprecated_models,
    is_model, list_modules, is_model_in_modules, is_model_pretrained, get_pretrained_cfg,
    get_pretrained_cfg_value, get_pretrained_cfgs_for_arch,
)

# architecture modules (registration happens at import time)
from .beit import *
from .byoanet import *
from .byobnet import *
from .cait import *
from .convmixer import *
from .convnext import *
from .densenet import *
from .davit import *
from .deit import *
from .dpn import *
from .edgenext import *
from .efficientformer import *
from .efficientnet import *
from .eva import *
from .focalnet import *
from .gcvit import *
from .gemma4_vit import *
from .ghostnet import *
from .hardcorenas import *
from .hiera import *
from .inception_next import *
from .inception_v3 import *
from .maxxvit import *
from .metaformer import *
from .mlp_mixer import *
from .mobilenetv3 import *
from .mobilenetv5 import *
from .mobilevit import *
from .nfnet import *
from .naflexvit import *
from .swin_transformer import *
from .swin_transformer_v2 import *
from .regnet import *
from .pit import *
from .pvt_v2 import *
from .res2net import *
from .resnest import *
from .resnetv2 import *
from .resnet import *
from .rexnet import *
from .selecsls import *
from .starnet import *
from .sknet import *
from .tiny_vit import *
from .tresnet import *
from .twins import *
from .vgg import *
from .xcit import *
from .vision_transformer import *
from .vovnet import *
from .xception import *
from .xception_aligned import *
from .cpubone import *
from .convit import *
from .senet import *
from .visformer import *
from .fastvit import *
from .fasternet import *
from .shvit import *
from .dla import *
from .csatv2 import *
from .cspnet import *
from .repvit import *
from .swiftformer import *
from .sequencer import *
from .repghost import *
from .rdnet import *
from .mambaout import *
from .crossvit import *
from .tnt import *
from .nest import *
from .levit import *
from .efficientformer_v2 import *
from .vitamin import *
from .volo import *
from .mvitv2 import *
from .hieradet_sam2 import *
from .hgnet import *
from .nextvit import *
from .coat import *
from .vision_transformer_relpos import *
from .efficientvit_msra import *
from .efficientvit_mit import *
from .hrnet import *
from .pnasnet import *
from .nasnet import *
from .inception_v4 import *
from .inception_resnet_v2 import *
from .vision_transformer_hybrid import *
from .vision_transformer_sam import *
from .swin_transformer_v2_cr import *
