"""EdgeNeXt — MI355X-native implementation.

Capability parity with reference `timm/models/edgenext.py`: `ConvBlock`
(:84), `CrossCovarianceAttn` (:141), `SplitTransposeBlock` (:183) with
multi-scale depthwise split + XCA, `EdgeNeXtStage` (:273), `EdgeNeXt` (:355)
and the xx_small..base (+_rw) variants.
"""
import math
from functools import partial
from typing import List, Optional, Tuple, Type, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..data.constants import IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD
from ..layers import (
    ClassifierHead, DropPath, LayerNorm2d, Mlp, NormMlpClassifierHead, create_conv2d, trunc_normal_,
)
from ._builder import build_model_with_cfg
from ._features import feature_take_indices
from ._features_fx import register_notrace_module
from ._manipulate import checkpoint_seq, named_apply
from ._registry import generate_default_cfgs, register_model

__all__ = ['EdgeNeXt']


@register_notrace_module  # reason: FX can't symbolically trace torch.arange in forward
class PositionalEncodingFourier(nn.Module):
    def __init__(self, hidden_dim: int = 32, dim: int = 768, temperature: float = 10000.):
        super().__init__()
        self.token_projection = nn.Conv2d(hidden_dim * 2, dim, kernel_size=1)
        self.scale = 2 * math.pi
        self.temperature = temperature
        self.hidden_dim = hidden_dim
        self.dim = dim

    def _axis_phase(self, n: int, device) -> torch.Tensor:
        """Normalized 1..n axis positions scaled to [~0, 2pi]."""
        coords = torch.arange(1, n + 1, dtype=torch.float32, device=device)
        return coords / (n + 1e-6) * self.scale

    def _sincos(self, phase: torch.Tensor, dim_t: torch.Tensor) -> torch.Tensor:
        # interleave sin/cos over frequency pairs -> [..., hidden_dim]
        freq = phase[..., None] / dim_t
        return torch.stack((freq[..., 0::2].sin(), freq[..., 1::2].cos()), dim=-1).flatten(-2)

    def forward(self, shape: Tuple[int, int, int]) -> torch.Tensor:
        B, H, W = shape
        device = self.token_projection.weight.device
        dtype = self.token_projection.weight.dtype
        dim_t = torch.arange(self.hidden_dim, dtype=torch.int64, device=device).to(torch.float32)
        dim_t = self.temperature ** (2 * torch.div(dim_t, 2, rounding_mode='floor') / self.hidden_dim)

        pos_y = self._sincos(
            self._axis_phase(H, device).view(1, H, 1).expand(B, H, W), dim_t)
        pos_x = self._sincos(
            self._axis_phase(W, device).view(1, 1, W).expand(B, H, W), dim_t)
        pos = torch.cat((pos_y, pos_x), dim=3).permute(0, 3, 1, 2)
        return self.token_projection(pos.to(dtype))


class ConvBlock(nn.Module):
    """ConvNeXt-style block w/ variable kernel size (reference `edgenext.py:84`)."""

    def __init__(
            self,
            dim: int,
            dim_out: Optional[int] = None,
            kernel_size: int = 7,
            stride: int = 1,
            conv_bias: bool = True,
            expand_ratio: float = 4,
            ls_init_value: float = 1e-6,
            norm_layer: Type[nn.Module] = partial(nn.LayerNorm, eps=1e-6),
            act_layer: Type[nn.Module] = nn.GELU,
            drop_path: float = 0.,
    ):
        super().__init__()
        dim_out = dim_out or dim
        self.shortcut_after_dw = stride > 1 or dim != dim_out

        self.conv_dw = create_conv2d(
            dim, dim_out, kernel_size=kernel_size, stride=stride, depthwise=True, bias=conv_bias)
        self.norm = norm_layer(dim_out)
        self.mlp = Mlp(dim_out, int(expand_ratio * dim_out), act_layer=act_layer)
        self.gamma = nn.Parameter(ls_init_value * torch.ones(dim_out)) if ls_init_value > 0 else None
        self.drop_path = DropPath(drop_path) if drop_path > 0. else nn.Identity()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shortcut = x
        x = self.conv_dw(x)
        if self.shortcut_after_dw:
            # stride/width change: residual taps the dw output instead
            shortcut = x
        return shortcut + self.drop_path(self._mlp_nchw(x))

    def _mlp_nchw(self, x: torch.Tensor) -> torch.Tensor:
        """norm + MLP + layer-scale computed channels-last, IO channels-first."""
        x = self.mlp(self.norm(x.permute(0, 2, 3, 1)))
        if self.gamma is not None:
            x = self.gamma * x
        return x.permute(0, 3, 1, 2)


class CrossCovarianceAttn(nn.Module):
    """XCA over channels (reference `edgenext.py:141`)."""

    def __init__(
            self,
            dim: int,
            num_heads: int = 8,
            qkv_bias: bool = False,
            attn_drop: float = 0.,
            proj_drop: float = 0.,
    ):
        super().__init__()
        self.num_heads = num_heads
        self.temperature = nn.Parameter(torch.ones(num_heads, 1, 1))

        self.qkv = nn.Linear(dim, dim * 3, bias=qkv_bias)
        self.attn_drop = nn.Dropout(attn_drop)
        self.proj = nn.Linear(dim, dim)
        self.proj_drop = nn.Dropout(proj_drop)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, N, C = x.shape
        qkv = self.qkv(x).reshape(B, N, 3, self.num_heads, -1).permute(2, 0, 3, 4, 1)
        q, k, v = qkv.unbind(0)

        # C x C attention map, not spatial
        attn = (F.normalize(q, dim=-1) @ F.normalize(k, dim=-1).transpose(-2, -1)) * self.temperature
        attn = attn.softmax(dim=-1)
        attn = self.attn_drop(attn)
        x = (attn @ v)

        x = x.permute(0, 3, 1, 2).reshape(B, N, C)
        x = self.proj(x)
        x = self.proj_drop(x)
        return x

    @torch.jit.ignore
    def no_weight_decay(self):
        return {'temperature'}


class SplitTransposeBlock(nn.Module):
    """Multi-scale depthwise split + XCA + MLP (reference `edgenext.py:183`)."""

    def __init__(
            self,
            dim: int,
            num_scales: int = 1,
            num_heads: int = 8,
            expand_ratio: float = 4,
            use_pos_emb: bool = True,
            conv_bias: bool = True,
            qkv_bias: bool = True,
            ls_init_value: float = 1e-6,
            norm_layer: Type[nn.Module] = partial(nn.LayerNorm, eps=1e-6),
            act_layer: Type[nn.Module] = nn.GELU,
            drop_path: float = 0.,
            attn_drop: float = 0.,
            proj_drop: float = 0.,
    ):
        super().__init__()
        width = max(int(math.ceil(dim / num_scales)), int(math.floor(dim // num_scales)))
        self.width = width
        self.num_scales = max(1, num_scales - 1)

        convs = []
        for i in range(self.num_scales):
            convs.append(create_conv2d(width, width, kernel_size=3, depthwise=True, bias=conv_bias))
        self.convs = nn.ModuleList(convs)

        self.pos_embd = None
        if use_pos_emb:
            self.pos_embd = PositionalEncodingFourier(dim=dim)
        self.norm_xca = norm_layer(dim)
        self.gamma_xca = nn.Parameter(ls_init_value * torch.ones(dim)) if ls_init_value > 0 else None
        self.xca = CrossCovarianceAttn(
            dim, num_heads=num_heads, qkv_bias=qkv_bias, attn_drop=attn_drop, proj_drop=proj_drop)

        self.norm = norm_layer(dim, eps=1e-6)
        self.mlp = Mlp(dim, int(expand_ratio * dim), act_layer=act_layer)
        self.gamma = nn.Parameter(ls_init_value * torch.ones(dim)) if ls_init_value > 0 else None
        self.drop_path = DropPath(drop_path) if drop_path > 0. else nn.Identity()

    def _multi_scale_mix(self, x: torch.Tensor) -> torch.Tensor:
        """Res2Net-style cascaded depthwise mixing over channel splits."""
        splits = x.chunk(len(self.convs) + 1, dim=1)
        outs = []
        acc = splits[0]
        for i, conv in enumerate(self.convs):
            if i > 0:
                acc = acc + splits[i]
            acc = conv(acc)
            outs.append(acc)
        outs.append(splits[-1])
        return torch.cat(outs, 1)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shortcut = x
        x = self._multi_scale_mix(x)

        # cross-covariance attention over flattened tokens
        B, C, H, W = x.shape
        tokens = x.reshape(B, C, H * W).permute(0, 2, 1)
        if self.pos_embd is not None:
            pos = self.pos_embd((B, H, W)).reshape(B, -1, tokens.shape[1]).permute(0, 2, 1)
            tokens = tokens + pos
        tokens = tokens + self.drop_path(self.gamma_xca * self.xca(self.norm_xca(tokens)))

        # inverted-bottleneck MLP, channels-last
        y = self.mlp(self.norm(tokens.reshape(B, H, W, C)))
        if self.gamma is not None:
            y = self.gamma * y
        return shortcut + self.drop_path(y.permute(0, 3, 1, 2))


class EdgeNeXtStage(nn.Module):
    def __init__(
            self,
            in_chs: int,
            out_chs: int,
            stride: int = 2,
            depth: int = 2,
            num_global_blocks: int = 1,
            num_heads: int = 4,
            scales: int = 2,
            kernel_size: int = 7,
            expand_ratio: float = 4,
            use_pos_emb: bool = False,
            downsample_block: bool = False,
            conv_bias: bool = True,
            ls_init_value: float = 1.0,
            drop_path_rates: Optional[List[float]] = None,
            norm_layer: Type[nn.Module] = LayerNorm2d,
            norm_layer_cl: Type[nn.Module] = partial(nn.LayerNorm, eps=1e-6),
            act_layer: Type[nn.Module] = nn.GELU,
    ):
        super().__init__()
        self.grad_checkpointing = False
        drop_path_rates = drop_path_rates or [0.] * depth

        if downsample_block or stride == 1:
            self.downsample = nn.Identity()
        else:
            self.downsample = nn.Sequential(
                norm_layer(in_chs),
                nn.Conv2d(in_chs, out_chs, kernel_size=2, stride=2, bias=conv_bias),
            )
            in_chs = out_chs

        # local ConvNeXt-style blocks first, global (XCA) blocks at the tail
        common = dict(
            expand_ratio=expand_ratio,
            conv_bias=conv_bias,
            ls_init_value=ls_init_value,
            norm_layer=norm_layer_cl,
            act_layer=act_layer,
        )
        stage_blocks = []
        for i in range(depth):
            if i < depth - num_global_blocks:
                block = ConvBlock(
                    dim=in_chs,
                    dim_out=out_chs,
                    stride=stride if downsample_block and i == 0 else 1,
                    kernel_size=kernel_size,
                    drop_path=drop_path_rates[i],
                    **common,
                )
            else:
                block = SplitTransposeBlock(
                    dim=in_chs,
                    num_scales=scales,
                    num_heads=num_heads,
                    use_pos_emb=use_pos_emb,
                    drop_path=drop_path_rates[i],
                    **common,
                )
            stage_blocks.append(block)
            in_chs = out_chs
        self.blocks = nn.Sequential(*stage_blocks)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.downsample(x)
        if self.grad_checkpointing and not torch.jit.is_scripting():
            x = checkpoint_seq(self.blocks, x)
        else:
            x = self.blocks(x)
        return x


class EdgeNeXt(nn.Module):
    """EdgeNeXt (reference `edgenext.py:355`; paper 2206.10589)."""

    def __init__(
            self,
            in_chans: int = 3,
            num_classes: int = 1000,
            global_pool: str = 'avg',
            dims: Tuple[int, ...] = (24, 48, 88, 168),
            depths: Tuple[int, ...] = (3, 3, 9, 3),
            global_block_counts: Tuple[int, ...] = (0, 1, 1, 1),
            kernel_sizes: Tuple[int, ...] = (3, 5, 7, 9),
            heads: Tuple[int, ...] = (8, 8, 8, 8),
            d2_scales: Tuple[int, ...] = (2, 2, 3, 4),
            use_pos_emb: Tuple[bool, ...] = (False, True, False, False),
            ls_init_value: float = 1e-6,
            head_init_scale: float = 1.,
            expand_ratio: float = 4,
            downsample_block: bool = False,
            conv_bias: bool = True,
            stem_type: str = 'patch',
            head_norm_first: bool = False,
            act_layer: Type[nn.Module] = nn.GELU,
            drop_path_rate: float = 0.,
            drop_rate: float = 0.,
    ):
        super().__init__()
        self.num_classes = num_classes
        self.global_pool = global_pool
        self.drop_rate = drop_rate
        norm_layer = partial(LayerNorm2d, eps=1e-6)
        norm_layer_cl = partial(nn.LayerNorm, eps=1e-6)
        self.feature_info = []

        assert stem_type in ('patch', 'overlap')
        if stem_type == 'patch':
            self.stem = nn.Sequential(
                nn.Conv2d(in_chans, dims[0], kernel_size=4, stride=4, bias=conv_bias),
                norm_layer(dims[0]),
            )
        else:
            self.stem = nn.Sequential(
                nn.Conv2d(in_chans, dims[0], kernel_size=9, stride=4, padding=9 // 2, bias=conv_bias),
                norm_layer(dims[0]),
            )

        curr_stride = 4
        stages = []
        dp_rates = [x.tolist() for x in torch.linspace(0, drop_path_rate, sum(depths)).split(depths)]
        in_chs = dims[0]
        for i in range(4):
            stride = 2 if curr_stride == 2 or i > 0 else 1
            curr_stride *= stride
            stages.append(EdgeNeXtStage(
                in_chs=in_chs,
                out_chs=dims[i],
                stride=stride,
                depth=depths[i],
                num_global_blocks=global_block_counts[i],
                num_heads=heads[i],
                drop_path_rates=dp_rates[i],
                scales=d2_scales[i],
                expand_ratio=expand_ratio,
                kernel_size=kernel_sizes[i],
                use_pos_emb=use_pos_emb[i],
                ls_init_value=ls_init_value,
                downsample_block=downsample_block,
                conv_bias=conv_bias,
                norm_layer=norm_layer,
                norm_layer_cl=norm_layer_cl,
                act_layer=act_layer,
            ))
            in_chs = dims[i]
            self.feature_info += [dict(num_chs=in_chs, reduction=curr_stride, module=f'stages.{i}')]

        self.stages = nn.Sequential(*stages)

        self.num_features = self.head_hidden_size = dims[-1]
        if head_norm_first:
            self.norm_pre = norm_layer(self.num_features)
            self.head = ClassifierHead(
                self.num_features, num_classes, pool_type=global_pool, drop_rate=self.drop_rate)
        else:
            self.norm_pre = nn.Identity()
            self.head = NormMlpClassifierHead(
                self.num_features, num_classes, pool_type=global_pool, drop_rate=self.drop_rate,
                norm_layer=norm_layer)

        named_apply(partial(_init_weights, head_init_scale=head_init_scale), self)

    @torch.jit.ignore
    def group_matcher(self, coarse: bool = False):
        return dict(
            stem=r'^stem',
            blocks=r'^stages\.(\d+)' if coarse else [
                (r'^stages\.(\d+)\.downsample', (0,)),
                (r'^stages\.(\d+)\.blocks\.(\d+)', None),
                (r'^norm_pre', (99999,))
            ]
        )

    @torch.jit.ignore
    def set_grad_checkpointing(self, enable: bool = True):
        for s in self.stages:
            s.grad_checkpointing = enable

    @torch.jit.ignore
    def get_classifier(self) -> nn.Module:
        return self.head.fc

    def reset_classifier(self, num_classes: int, global_pool: Optional[str] = None):
        self.num_classes = num_classes
        self.head.reset(num_classes, global_pool)

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

        x = self.stem(x)
        if torch.jit.is_scripting() or not stop_early:
            stages = self.stages
        else:
            stages = self.stages[:max_index + 1]
        for feat_idx, stage in enumerate(stages):
            x = stage(x)
            if feat_idx in take_indices:
                intermediates.append(x)

        if intermediates_only:
            return intermediates
        x = self.norm_pre(x)
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
            self.norm_pre = nn.Identity()
        if prune_head:
            self.reset_classifier(0, '')
        return take_indices

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        x = self.stem(x)
        x = self.stages(x)
        x = self.norm_pre(x)
        return x

    def forward_head(self, x: torch.Tensor, pre_logits: bool = False) -> torch.Tensor:
        return self.head(x, pre_logits=pre_logits) if pre_logits else self.head(x)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.forward_features(x)
        x = self.forward_head(x)
        return x


def _init_weights(module: nn.Module, name: Optional[str] = None, head_init_scale: float = 1.0):
    if isinstance(module, nn.Conv2d):
        trunc_normal_(module.weight, std=.02)
        if module.bias is not None:
            nn.init.zeros_(module.bias)
    elif isinstance(module, nn.Linear):
        trunc_normal_(module.weight, std=.02)
        nn.init.zeros_(module.bias)
        if name and 'head.' in name:
            module.weight.data.mul_(head_init_scale)
            module.bias.data.mul_(head_init_scale)


def _create_edgenext(variant, pretrained=False, **kwargs):
    model = build_model_with_cfg(
        EdgeNeXt, variant, pretrained,
        feature_cfg=dict(out_indices=(0, 1, 2, 3), flatten_sequential=True),
        **kwargs,
    )
    return model


def _cfg(url='', **kwargs):
    return {
        'url': url, 'num_classes': 1000, 'input_size': (3, 256, 256), 'pool_size': (8, 8),
        'crop_pct': 0.9, 'interpolation': 'bicubic',
        'mean': IMAGENET_DEFAULT_MEAN, 'std': IMAGENET_DEFAULT_STD,
        'first_conv': 'stem.0', 'classifier': 'head.fc',
        **kwargs,
    }


default_cfgs = generate_default_cfgs({
    'edgenext_xx_small.in1k': _cfg(test_input_size=(3, 288, 288), test_crop_pct=1.0),
    'edgenext_x_small.in1k': _cfg(test_input_size=(3, 288, 288), test_crop_pct=1.0),
    'edgenext_small.usi_in1k': _cfg(crop_pct=0.95, test_input_size=(3, 320, 320), test_crop_pct=1.0),
    'edgenext_base.usi_in1k': _cfg(crop_pct=0.95, test_input_size=(3, 320, 320), test_crop_pct=1.0),
    'edgenext_base.in21k_ft_in1k': _cfg(crop_pct=0.95, test_input_size=(3, 320, 320), test_crop_pct=1.0),
    'edgenext_small_rw.sw_in1k': _cfg(test_input_size=(3, 320, 320), test_crop_pct=1.0),
})


@register_model
def edgenext_xx_small(pretrained=False, **kwargs) -> EdgeNeXt:
    model_args = dict(depths=(2, 2, 6, 2), dims=(24, 48, 88, 168), heads=(4, 4, 4, 4))
    return _create_edgenext('edgenext_xx_small', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def edgenext_x_small(pretrained=False, **kwargs) -> EdgeNeXt:
    model_args = dict(depths=(3, 3, 9, 3), dims=(32, 64, 100, 192), heads=(4, 4, 4, 4))
    return _create_edgenext('edgenext_x_small', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def edgenext_small(pretrained=False, **kwargs) -> EdgeNeXt:
    model_args = dict(depths=(3, 3, 9, 3), dims=(48, 96, 160, 304))
    return _create_edgenext('edgenext_small', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def edgenext_base(pretrained=False, **kwargs) -> EdgeNeXt:
    model_args = dict(depths=[3, 3, 9, 3], dims=[80, 160, 288, 584])
    return _create_edgenext('edgenext_base', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def edgenext_small_rw(pretrained=False, **kwargs) -> EdgeNeXt:
    model_args = dict(
        depths=(3, 3, 9, 3), dims=(48, 96, 192, 384),
        downsample_block=True, conv_bias=False, stem_type='overlap')
    return _create_edgenext('edgenext_small_rw', pretrained=pretrained, **dict(model_args, **kwargs))
