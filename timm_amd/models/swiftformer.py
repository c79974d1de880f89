"""SwiftFormer — MI355X-native implementation.

Capability parity with reference `timm/models/swiftformer.py`: efficient
additive attention (global query vector via learned `w_g`, :154), conv
encoders for all but the last block of each stage (:282), dual
classification + distillation heads averaged at inference.
"""
import re
from typing import Any, Dict, List, Optional, Tuple, Type, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..data.constants import IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD
from ..layers import DropPath, to_2tuple, trunc_normal_
from ._builder import build_model_with_cfg
from ._features import feature_take_indices
from ._manipulate import checkpoint_seq
from ._registry import generate_default_cfgs, register_model

__all__ = ['SwiftFormer']


class LayerScale2d(nn.Module):
    def __init__(self, dim: int, init_values: float = 1e-5, inplace: bool = False):
        super().__init__()
        self.inplace = inplace
        self.gamma = nn.Parameter(init_values * torch.ones(dim, 1, 1))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x.mul_(self.gamma) if self.inplace else x * self.gamma


class Embedding(nn.Module):
    """Conv patch embedding between stages."""

    def __init__(self, in_chans=3, embed_dim=768, patch_size=16, stride=16, padding=0,
                 norm_layer=nn.BatchNorm2d):
        super().__init__()
        patch_size = to_2tuple(patch_size)
        stride = to_2tuple(stride)
        padding = to_2tuple(padding)
        self.proj = nn.Conv2d(in_chans, embed_dim, patch_size, stride, padding)
        self.norm = norm_layer(embed_dim) if norm_layer else nn.Identity()

    def forward(self, x):
        return self.norm(self.proj(x))


class ConvEncoder(nn.Module):
    def __init__(self, dim, hidden_dim=64, kernel_size=3, drop_path=0.,
                 act_layer=nn.GELU, norm_layer=nn.BatchNorm2d, use_layer_scale=True):
        super().__init__()
        self.dwconv = nn.Conv2d(dim, dim, kernel_size, padding=kernel_size // 2, groups=dim)
        self.norm = norm_layer(dim)
        self.pwconv1 = nn.Conv2d(dim, hidden_dim, 1)
        self.act = act_layer()
        self.pwconv2 = nn.Conv2d(hidden_dim, dim, 1)
        self.drop_path = DropPath(drop_path) if drop_path > 0. else nn.Identity()
        self.layer_scale = LayerScale2d(dim, 1) if use_layer_scale else nn.Identity()

    def forward(self, x):
        inp = x
        x = self.norm(self.dwconv(x))
        x = self.pwconv2(self.act(self.pwconv1(x)))
        x = self.layer_scale(x)
        return inp + self.drop_path(x)


class ConvMlp(nn.Module):
    def __init__(self, in_features, hidden_features=None, out_features=None,
                 act_layer=nn.GELU, norm_layer=nn.BatchNorm2d, drop=0.):
        super().__init__()
        out_features = out_features or in_features
        hidden_features = hidden_features or in_features
        self.norm1 = norm_layer(in_features)
        self.fc1 = nn.Conv2d(in_features, hidden_features, 1)
        self.act = act_layer()
        self.fc2 = nn.Conv2d(hidden_features, out_features, 1)
        self.drop = nn.Dropout(drop)

    def forward(self, x):
        x = self.norm1(x)
        x = self.drop(self.act(self.fc1(x)))
        x = self.drop(self.fc2(x))
        return x


class EfficientAdditiveAttention(nn.Module):
    """Additive attention: global query vector instead of full QK matrix."""

    def __init__(self, in_dims=512, token_dim=256, num_heads=1):
        super().__init__()
        self.scale_factor = token_dim ** -0.5
        self.to_query = nn.Linear(in_dims, token_dim * num_heads)
        self.to_key = nn.Linear(in_dims, token_dim * num_heads)
        self.w_g = nn.Parameter(torch.randn(token_dim * num_heads, 1))
        self.proj = nn.Linear(token_dim * num_heads, token_dim * num_heads)
        self.final = nn.Linear(token_dim * num_heads, token_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, _, H, W = x.shape
        x = x.flatten(2).permute(0, 2, 1)

        query = F.normalize(self.to_query(x), dim=-1)
        key = F.normalize(self.to_key(x), dim=-1)

        attn = F.normalize(query @ self.w_g * self.scale_factor, dim=1)
        attn = torch.sum(attn * query, dim=1, keepdim=True)

        out = self.proj(attn * key) + query
        out = self.final(out).permute(0, 2, 1).reshape(B, -1, H, W)
        return out


class LocalRepresentation(nn.Module):
    def __init__(self, dim, kernel_size=3, drop_path=0., use_layer_scale=True,
                 act_layer=nn.GELU, norm_layer=nn.BatchNorm2d):
        super().__init__()
        self.dwconv = nn.Conv2d(dim, dim, kernel_size, padding=kernel_size // 2, groups=dim)
        self.norm = norm_layer(dim)
        self.pwconv1 = nn.Conv2d(dim, dim, kernel_size=1)
        self.act = act_layer()
        self.pwconv2 = nn.Conv2d(dim, dim, kernel_size=1)
        self.drop_path = DropPath(drop_path) if drop_path > 0. else nn.Identity()
        self.layer_scale = LayerScale2d(dim, 1) if use_layer_scale else nn.Identity()

    def forward(self, x):
        skip = x
        x = self.norm(self.dwconv(x))
        x = self.pwconv2(self.act(self.pwconv1(x)))
        x = self.layer_scale(x)
        return skip + self.drop_path(x)


class Block(nn.Module):
    """Local rep → efficient additive attention → conv MLP."""

    def __init__(self, dim, mlp_ratio=4., drop_rate=0., drop_path=0.,
                 act_layer=nn.GELU, norm_layer=nn.BatchNorm2d,
                 use_layer_scale=True, layer_scale_init_value=1e-5):
        super().__init__()
        self.local_representation = LocalRepresentation(
            dim=dim, use_layer_scale=use_layer_scale, act_layer=act_layer, norm_layer=norm_layer)
        self.attn = EfficientAdditiveAttention(in_dims=dim, token_dim=dim)
        self.linear = ConvMlp(
            in_features=dim, hidden_features=int(dim * mlp_ratio),
            act_layer=act_layer, norm_layer=norm_layer, drop=drop_rate)
        self.drop_path = DropPath(drop_path) if drop_path > 0. else nn.Identity()
        self.layer_scale_1 = LayerScale2d(dim, layer_scale_init_value) if use_layer_scale else nn.Identity()
        self.layer_scale_2 = LayerScale2d(dim, layer_scale_init_value) if use_layer_scale else nn.Identity()

    def forward(self, x):
        x = self.local_representation(x)
        x = x + self.drop_path(self.layer_scale_1(self.attn(x)))
        x = x + self.drop_path(self.layer_scale_2(self.linear(x)))
        return x


class Stage(nn.Module):
    def __init__(self, dim, index, layers, mlp_ratio=4., act_layer=nn.GELU,
                 norm_layer=nn.BatchNorm2d, drop_rate=0., drop_path_rate=0.,
                 use_layer_scale=True, layer_scale_init_value=1e-5, downsample=None):
        super().__init__()
        self.grad_checkpointing = False
        self.downsample = downsample if downsample is not None else nn.Identity()

        blocks = []
        for block_idx in range(layers[index]):
            block_dpr = drop_path_rate * (block_idx + sum(layers[:index])) / (sum(layers) - 1)
            if layers[index] - block_idx <= 1:
                blocks.append(Block(
                    dim, mlp_ratio=mlp_ratio, drop_rate=drop_rate, drop_path=block_dpr,
                    act_layer=act_layer, norm_layer=norm_layer,
                    use_layer_scale=use_layer_scale, layer_scale_init_value=layer_scale_init_value))
            else:
                blocks.append(ConvEncoder(
                    dim=dim, hidden_dim=int(mlp_ratio * dim), kernel_size=3, drop_path=block_dpr,
                    act_layer=act_layer, norm_layer=norm_layer, use_layer_scale=use_layer_scale))
        self.blocks = nn.Sequential(*blocks)

    def forward(self, x):
        x = self.downsample(x)
        if self.grad_checkpointing and not torch.jit.is_scripting():
            x = checkpoint_seq(self.blocks, x)
        else:
            x = self.blocks(x)
        return x


class SwiftFormer(nn.Module):
    """SwiftFormer (reference `swiftformer.py:346`; paper 2303.15446)."""

    def __init__(
            self,
            layers: Tuple[int, ...] = (3, 3, 6, 4),
            embed_dims: Tuple[int, ...] = (48, 56, 112, 220),
            mlp_ratios: float = 4,
            downsamples: Tuple[bool, ...] = (False, True, True, True),
            act_layer: Type[nn.Module] = nn.GELU,
            down_patch_size: int = 3,
            down_stride: int = 2,
            down_pad: int = 1,
            num_classes: int = 1000,
            drop_rate: float = 0.,
            drop_path_rate: float = 0.,
            use_layer_scale: bool = True,
            layer_scale_init_value: float = 1e-5,
            global_pool: str = 'avg',
            output_stride: int = 32,
            in_chans: int = 3,
            **kwargs,
    ):
        super().__init__()
        assert output_stride == 32
        self.num_classes = num_classes
        self.global_pool = global_pool
        self.feature_info = []

        self.stem = nn.Sequential(
            nn.Conv2d(in_chans, embed_dims[0] // 2, 3, 2, 1),
            nn.BatchNorm2d(embed_dims[0] // 2),
            nn.ReLU(),
            nn.Conv2d(embed_dims[0] // 2, embed_dims[0], 3, 2, 1),
            nn.BatchNorm2d(embed_dims[0]),
            nn.ReLU(),
        )
        prev_dim = embed_dims[0]

        stages = []
        for i in range(len(layers)):
            downsample = Embedding(
                in_chans=prev_dim, embed_dim=embed_dims[i], patch_size=down_patch_size,
                stride=down_stride, padding=down_pad) if downsamples[i] else nn.Identity()
            stage = Stage(
                dim=embed_dims[i], index=i, layers=layers, mlp_ratio=mlp_ratios,
                act_layer=act_layer, drop_rate=drop_rate, drop_path_rate=drop_path_rate,
                use_layer_scale=use_layer_scale, layer_scale_init_value=layer_scale_init_value,
                downsample=downsample)
            prev_dim = embed_dims[i]
            stages.append(stage)
            self.feature_info += [dict(num_chs=embed_dims[i], reduction=2 ** (i + 2), module=f'stages.{i}')]
        self.stages = nn.Sequential(*stages)

        self.num_features = self.head_hidden_size = out_chs = embed_dims[-1]
        self.norm = nn.BatchNorm2d(out_chs)
        self.head_drop = nn.Dropout(drop_rate)
        self.head = nn.Linear(out_chs, num_classes) if num_classes > 0 else nn.Identity()
        self.head_dist = nn.Linear(out_chs, num_classes) if num_classes > 0 else nn.Identity()
        self.distilled_training = False
        self._initialize_weights()

    def _initialize_weights(self):
        for name, m in self.named_modules():
            if isinstance(m, (nn.Linear, nn.Conv2d)):
                trunc_normal_(m.weight, std=.02)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)

    @torch.jit.ignore
    def no_weight_decay(self):
        return set()

    @torch.jit.ignore
    def group_matcher(self, coarse: bool = False):
        return dict(
            stem=r'^stem',
            blocks=r'^stages\.(\d+)' if coarse else [
                (r'^stages\.(\d+).downsample', (0,)),
                (r'^stages\.(\d+)\.blocks\.(\d+)', None),
                (r'^norm', (99999,)),
            ]
        )

    @torch.jit.ignore
    def set_grad_checkpointing(self, enable: bool = True):
        for s in self.stages:
            s.grad_checkpointing = enable

    @torch.jit.ignore
    def get_classifier(self) -> Tuple[nn.Module, nn.Module]:
        return self.head, self.head_dist

    def reset_classifier(self, num_classes: int, global_pool: Optional[str] = None):
        self.num_classes = num_classes
        if global_pool is not None:
            self.global_pool = global_pool
        self.head = nn.Linear(self.num_features, num_classes) if num_classes > 0 else nn.Identity()
        self.head_dist = nn.Linear(self.num_features, num_classes) if num_classes > 0 else nn.Identity()

    @torch.jit.ignore
    def set_distilled_training(self, enable: bool = True):
        self.distilled_training = enable

    def forward_intermediates(
            self,
            x: torch.Tensor,
            indices: Optional[Union[int, List[int]]] = None,
            norm: bool = False,
            stop_early: bool = False,
            output_fmt: str = 'NCHW',
            intermediates_only: bool = False,
    ) -> Union[List[torch.Tensor], Tuple[torch.Tensor, List[torch.Tensor]]]:
        assert output_fmt in ('NCHW',), 'Output shape must be NCHW.'
        intermediates = []
        take_indices, max_index = feature_take_indices(len(self.stages), indices)
        last_idx = len(self.stages) - 1

        x = self.stem(x)
        if torch.jit.is_scripting() or not stop_early:
            stages = self.stages
        else:
            stages = self.stages[:max_index + 1]

        for feat_idx, stage in enumerate(stages):
            x = stage(x)
            if feat_idx in take_indices:
                x_inter = self.norm(x) if norm and feat_idx == last_idx else x
                intermediates.append(x_inter)

        if intermediates_only:
            return intermediates

        if feat_idx == last_idx:
            x = self.norm(x)
        return x, intermediates

    def prune_intermediate_layers(
            self,
            indices: Union[int, List[int]] = 1,
            prune_norm: bool = False,
            prune_head: bool = True,
    ):
        take_indices, max_index = feature_take_indices(len(self.stages), indices)
        self.stages = self.stages[:max_index + 1]
        if prune_norm:
            self.norm = nn.Identity()
        if prune_head:
            self.reset_classifier(0, '')
        return take_indices

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        x = self.stem(x)
        x = self.stages(x)
        x = self.norm(x)
        return x

    def forward_head(self, x: torch.Tensor, pre_logits: bool = False):
        if self.global_pool == 'avg':
            x = x.mean(dim=(2, 3))
        x = self.head_drop(x)
        if pre_logits:
            return x
        x, x_dist = self.head(x), self.head_dist(x)
        if self.distilled_training and self.training and not torch.jit.is_scripting():
            return x, x_dist
        return (x + x_dist) / 2

    def forward(self, x: torch.Tensor):
        x = self.forward_features(x)
        x = self.forward_head(x)
        return x


def checkpoint_filter_fn(state_dict: Dict[str, torch.Tensor], model: nn.Module) -> Dict[str, torch.Tensor]:
    state_dict = state_dict.get('model', state_dict)
    if 'stem.0.weight' in state_dict:
        return state_dict
    out_dict = {}
    for k, v in state_dict.items():
        k = k.replace('patch_embed.', 'stem.')
        k = k.replace('dist_head.', 'head_dist.')
        k = k.replace('attn.Proj.', 'attn.proj.')
        k = k.replace('.layer_scale_1', '.layer_scale_1.gamma')
        k = k.replace('.layer_scale_2', '.layer_scale_2.gamma')
        k = re.sub(r'\.layer_scale(?=$|\.)', '.layer_scale.gamma', k)
        m = re.match(r'^network\.(\d+)\.(.*)', k)
        if m:
            n_idx, rest = int(m.group(1)), m.group(2)
            stage_idx = n_idx // 2
            if n_idx % 2 == 0:
                k = f'stages.{stage_idx}.blocks.{rest}'
            else:
                k = f'stages.{stage_idx + 1}.downsample.{rest}'
        out_dict[k] = v
    return out_dict


def _cfg(url: str = '', **kwargs: Any) -> Dict[str, Any]:
    return {
        'url': url, 'num_classes': 1000, 'input_size': (3, 224, 224), 'pool_size': None,
        'fixed_input_size': True, 'crop_pct': .95, 'interpolation': 'bicubic',
        'mean': IMAGENET_DEFAULT_MEAN, 'std': IMAGENET_DEFAULT_STD,
        'first_conv': 'stem.0', 'classifier': ('head', 'head_dist'),
        **kwargs,
    }


default_cfgs = generate_default_cfgs({
    'swiftformer_xs.dist_in1k': _cfg(),
    'swiftformer_s.dist_in1k': _cfg(),
    'swiftformer_l1.dist_in1k': _cfg(),
    'swiftformer_l3.dist_in1k': _cfg(),
})


def _create_swiftformer(variant: str, pretrained: bool = False, **kwargs: Any) -> SwiftFormer:
    return build_model_with_cfg(
        SwiftFormer, variant, pretrained,
        pretrained_filter_fn=checkpoint_filter_fn,
        feature_cfg=dict(out_indices=(0, 1, 2, 3), flatten_sequential=True),
        **kwargs,
    )


@register_model
def swiftformer_xs(pretrained: bool = False, **kwargs: Any) -> SwiftFormer:
    model_args = dict(layers=[3, 3, 6, 4], embed_dims=[48, 56, 112, 220])
    return _create_swiftformer('swiftformer_xs', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def swiftformer_s(pretrained: bool = False, **kwargs: Any) -> SwiftFormer:
    model_args = dict(layers=[3, 3, 9, 6], embed_dims=[48, 64, 168, 224])
    return _create_swiftformer('swiftformer_s', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def swiftformer_l1(pretrained: bool = False, **kwargs: Any) -> SwiftFormer:
    model_args = dict(layers=[4, 3, 10, 5], embed_dims=[48, 96, 192, 384])
    return _create_swiftformer('swiftformer_l1', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def swiftformer_l3(pretrained: bool = False, **kwargs: Any) -> SwiftFormer:
    model_args = dict(layers=[4, 4, 12, 6], embed_dims=[64, 128, 320, 512])
    return _create_swiftformer('swiftformer_l3', pretrained=pretrained, **dict(model_args, **kwargs))
