from .activations import *
from .adaptive_avgmax_pool import (
    AdaptiveAvgMaxPool2d, AdaptiveCatAvgMaxPool2d, SelectAdaptivePool2d,
    adaptive_avgmax_pool2d, adaptive_catavgmax_pool2d, adaptive_pool_feat_mult, select_adaptive_pool2d,
)
from .attention import Attention, AttentionRope, maybe_add_mask
from .attention2d import Attention2d, MultiQueryAttention2d, MultiQueryAttentionV2
from .attention_pool2d import AttentionPool2d, RotAttentionPool2d
from .attention_pool import AttentionPoolLatent
from .blur_pool import BlurPool2d, create_aa
from .classifier import ClassifierHead, ClNormMlpClassifierHead, NormMlpClassifierHead, create_classifier
from .cond_conv2d import CondConv2d, get_condconv_initializer
from .config import (
    is_exportable, is_no_jit, is_scriptable, set_exportable, set_layer_config, set_no_jit, set_scriptable,
    set_fused_attn, use_fused_attn, set_reentrant_ckpt, use_reentrant_ckpt,
)
from .conv2d_same import Conv2dSame, Conv2dSameExport, conv2d_same
from .conv_bn_act import ConvBnAct, ConvNormAct, ConvNormActAa
from .conv_helpers_extra import SeparableConv2d, SeparableConvBnAct, SeparableConvNormAct
from .create_act import create_act_layer, get_act_fn, get_act_layer
from .create_attn import create_attn, create_attn_layer, get_attn
from .create_conv2d import create_conv2d
from .eca import CecaModule, EcaModule, EfficientChannelAttn
from .gather_excite import GatherExcite
from .global_context import GlobalContext
from .create_norm import create_norm_layer, get_norm_layer
from .diff_attention import DiffAttention
from .drop import DropBlock2d, DropPath, calculate_drop_path_rates, drop_block_2d, drop_path
from .format import Format, FormatT, get_channel_dim, get_spatial_dim, nchw_to, nhwc_to
from .grn import GlobalResponseNorm
from .helpers import extend_tuple, make_divisible, to_2tuple, to_3tuple, to_4tuple, to_ntuple
from .layer_scale import LayerScale, LayerScale2d
from .linear import Linear
from .mixed_conv2d import MixedConv2d
from .mlp import ConvMlp, GatedMlp, GlobalResponseNormMlp, GluMlp, Mlp, SwiGLU, SwiGLUPacked
from .norm import (
    GroupNorm, GroupNorm1, LayerNorm, LayerNorm2d, LayerNormFp32, RmsNorm, RmsNorm2d, RmsNormFp32,
    SimpleNorm, SimpleNorm2d,
)
from .norm_act import (
    BatchNormAct2d, FrozenBatchNormAct2d, GroupNormAct, LayerNormAct, LayerNormAct2d, RmsNormAct, RmsNormAct2d, SyncBatchNormAct,
    convert_sync_batchnorm, create_norm_act_layer, freeze_batch_norm_2d, get_norm_act_layer,
    unfreeze_batch_norm_2d,
)
from .padding import get_padding, get_same_padding, pad_same
from .patch_dropout import PatchDropout, patch_dropout_forward
from .patch_embed import PatchEmbed, PatchEmbedWithSize, resample_patch_embed
from .hybrid_embed import HybridEmbed, HybridEmbedWithSize
from .pool2d_same import AvgPool2dSame, MaxPool2dSame, create_pool2d
from .pos_embed_rel import (
    RelPosBias, RelPosBiasTf, RelPosMlp, gen_relative_position_index,
    resize_rel_pos_bias_table, resize_rel_pos_bias_table_simple,
)
from .pos_embed import resample_abs_pos_embed, resample_abs_pos_embed_nhwc
from .pos_embed_sincos import (
    FourierEmbed, RotaryEmbedding, RotaryEmbeddingCat, apply_keep_indices_nlc, apply_rot_embed,
    apply_rot_embed_cat, apply_rot_embed_list, build_fourier_pos_embed, build_rotary_pos_embed,
    build_sincos2d_pos_embed, create_rope_embed, freq_bands, pixel_freq_bands, rope_rotate_half, rot,
)
from .std_conv import ScaledStdConv2d, ScaledStdConv2dSame, StdConv2d, StdConv2dSame
from .evo_norm import (
    EvoNorm2dB0, EvoNorm2dB1, EvoNorm2dB2, EvoNorm2dS0, EvoNorm2dS0a, EvoNorm2dS1, EvoNorm2dS1a,
    EvoNorm2dS2, EvoNorm2dS2a,
)
from .filter_response_norm import FilterResponseNormAct2d, FilterResponseNormTlu2d
from .cbam import CbamModule, LightCbamModule, ChannelAttn, LightChannelAttn, SpatialAttn, LightSpatialAttn
from .coord_attn import CoordAttn, EfficientLocalAttn, SimpleCoordAttn, StripAttn
from .conv_self_attn import BottleneckAttn, HaloAttn, LambdaLayer, RelPos2d
from .grid import ndgrid, meshgrid
from .non_local_attn import NonLocalAttn, BatNonLocalAttn
from .selective_kernel import SelectiveKernel, SelectiveKernelAttn
from .space_to_depth import DepthToSpace, SpaceToDepth
from .split_attn import RadixSoftmax, SplitAttn
from .split_batchnorm import SplitBatchNorm2d, convert_splitbn_model
from .squeeze_excite import EffectiveSEModule, EffectiveSqueezeExcite, SEModule, SqueezeExcite, SqueezeExciteCl
from .test_time_pool import TestTimePoolHead, apply_test_time_pool
from .trace_utils import _assert
from .weight_init import (
    init_weight_jax, init_weight_vit, lecun_normal_, trunc_normal_, trunc_normal_tf_, variance_scaling_,
)
from .activations_me import HardMishMe, HardSigmoidMe, HardSwishMe, MishMe, SwishMe
from .fast_norm import (
    fast_group_norm, fast_layer_norm, fast_rms_norm, is_fast_norm, set_fast_norm,
)
from .interpolate import RegularGridInterpolator
from .median_pool import MedianPool2d
from .ml_decoder import MLDecoder, add_ml_decoder_head
from .pool1d import global_pool_nlc
from .inplace_abn import InplaceAbn
from .typing import LayerType, PadType, disable_compiler, nullwrap
