"""CaiT (Class-Attention in Image Transformers) — MI355X-native implementation.

Capability parity with reference `timm/models/cait.py`: `ClassAttn` (:27) —
query computed for the cls token only — `TalkingHeadAttn` (:132) with
pre/post-softmax head mixing, `LayerScaleBlock[ClassAttn]` (:81/:184) and
`Cait` (:234).

Self-attention in the main blocks goes through the fused gfx950 flash kernel;
TalkingHeadAttn keeps the materialised-score composition (head-mixing linear
layers act on the N x N score tensor, which cannot be fused into an online
softmax).
"""
from functools import partial
from typing import Any, Callable, Dict, List, Optional, Tuple, Type, Union

import torch
import torch.nn as nn

from .. import ops
from ..data.constants import IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD
from ..layers import DropPath, LayerNorm, Mlp, PatchEmbed, trunc_normal_
from ._builder import build_model_with_cfg
from ._features import feature_take_indices
from ._manipulate import checkpoint, checkpoint_seq
from ._registry import generate_default_cfgs, register_model

__all__ = ['Cait', 'ClassAttn', 'LayerScaleBlock', 'LayerScaleBlockClassAttn', 'TalkingHeadAttn']


class ClassAttn(nn.Module):
    """Class attention: q from cls token only, k/v from all tokens (reference `cait.py:27`)."""

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
        head_dim = dim // num_heads
        self.scale = head_dim ** -0.5

        self.q = nn.Linear(dim, dim, bias=qkv_bias)
        self.k = nn.Linear(dim, dim, bias=qkv_bias)
        self.v = nn.Linear(dim, dim, bias=qkv_bias)
        self.attn_drop = nn.Dropout(attn_drop)
        self.proj = nn.Linear(dim, dim)
        self.proj_drop = nn.Dropout(proj_drop)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, N, C = x.shape
        q = self.q(x[:, 0]).unsqueeze(1).reshape(B, 1, self.num_heads, C // self.num_heads).permute(0, 2, 1, 3)
        k = self.k(x).reshape(B, N, self.num_heads, C // self.num_heads).permute(0, 2, 1, 3)
        v = self.v(x).reshape(B, N, self.num_heads, C // self.num_heads).permute(0, 2, 1, 3)

        x_cls = ops.flash_attention(
            q, k, v,
            dropout_p=self.attn_drop.p if self.training else 0.,
            scale=self.scale,
        )

        x_cls = x_cls.transpose(1, 2).reshape(B, 1, C)
        x_cls = self.proj(x_cls)
        x_cls = self.proj_drop(x_cls)
        return x_cls


class _LayerScaleBlockBase(nn.Module):
    """Shared norm/attn/mlp + per-branch gamma scaffolding for both CaiT block
    flavours; subclasses differ only in forward wiring + default attn."""

    def __init__(
            self,
            dim: int,
            num_heads: int,
            attn_block: Type[nn.Module],
            mlp_ratio: float = 4.,
            qkv_bias: bool = False,
            proj_drop: float = 0.,
            attn_drop: float = 0.,
            drop_path: float = 0.,
            act_layer: Type[nn.Module] = nn.GELU,
            norm_layer: Type[nn.Module] = LayerNorm,
            mlp_block: Type[nn.Module] = Mlp,
            init_values: float = 1e-4,
    ):
        super().__init__()
        self.norm1 = norm_layer(dim)
        self.attn = attn_block(
            dim, num_heads=num_heads, qkv_bias=qkv_bias, attn_drop=attn_drop, proj_drop=proj_drop)
        self.drop_path = DropPath(drop_path) if drop_path > 0. else nn.Identity()
        self.norm2 = norm_layer(dim)
        self.mlp = mlp_block(
            in_features=dim, hidden_features=int(dim * mlp_ratio), act_layer=act_layer, drop=proj_drop)
        self.gamma_1 = nn.Parameter(init_values * torch.ones(dim))
        self.gamma_2 = nn.Parameter(init_values * torch.ones(dim))


class LayerScaleBlockClassAttn(_LayerScaleBlockBase):
    """Class-attention block: only the cls token is updated (reference `cait.py:81`)."""

    def __init__(self, dim, num_heads, attn_block: Type[nn.Module] = ClassAttn, **kwargs):
        super().__init__(dim, num_heads, attn_block, **kwargs)

    def forward(self, x: torch.Tensor, x_cls: torch.Tensor) -> torch.Tensor:
        joint = torch.cat((x_cls, x), dim=1)
        x_cls = x_cls + self.drop_path(self.gamma_1 * self.attn(self.norm1(joint)))
        x_cls = x_cls + self.drop_path(self.gamma_2 * self.mlp(self.norm2(x_cls)))
        return x_cls


class TalkingHeadAttn(nn.Module):
    """Talking-heads attention: linear head mixing pre/post softmax (reference `cait.py:132`)."""

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
        head_dim = dim // num_heads
        self.scale = head_dim ** -0.5

        self.qkv = nn.Linear(dim, dim * 3, bias=qkv_bias)
        self.attn_drop = nn.Dropout(attn_drop)
        self.proj = nn.Linear(dim, dim)
        self.proj_l = nn.Linear(num_heads, num_heads)
        self.proj_w = nn.Linear(num_heads, num_heads)
        self.proj_drop = nn.Dropout(proj_drop)

    @staticmethod
    def _mix_heads(scores: torch.Tensor, mixer: nn.Linear) -> torch.Tensor:
        # linear blend across the head dim of the [B,H,N,N] score tensor
        return mixer(scores.permute(0, 2, 3, 1)).permute(0, 3, 1, 2)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, N, C = x.shape
        qkv = self.qkv(x).reshape(B, N, 3, self.num_heads, C // self.num_heads).permute(2, 0, 3, 1, 4)
        q, k, v = qkv[0] * self.scale, qkv[1], qkv[2]

        # scores stay materialised: the talking-heads mixers act on [B,H,N,N]
        # both sides of the softmax, which precludes an online-softmax kernel
        scores = self._mix_heads(q @ k.transpose(-2, -1), self.proj_l)
        scores = self._mix_heads(scores.softmax(dim=-1), self.proj_w)
        scores = self.attn_drop(scores)

        x = (scores @ v).transpose(1, 2).reshape(B, N, C)
        return self.proj_drop(self.proj(x))


class LayerScaleBlock(_LayerScaleBlockBase):
    """Self-attention block with per-branch LayerScale (reference `cait.py:184`)."""

    def __init__(self, dim, num_heads, attn_block: Type[nn.Module] = TalkingHeadAttn, **kwargs):
        super().__init__(dim, num_heads, attn_block, **kwargs)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x + self.drop_path(self.gamma_1 * self.attn(self.norm1(x)))
        x = x + self.drop_path(self.gamma_2 * self.mlp(self.norm2(x)))
        return x


class Cait(nn.Module):
    """CaiT model (reference `cait.py:234`; paper: Going deeper with Image Transformers)."""

    def __init__(
            self,
            img_size: Union[int, Tuple[int, int]] = 224,
            patch_size: int = 16,
            in_chans: int = 3,
            num_classes: int = 1000,
            global_pool: str = 'token',
            embed_dim: int = 768,
            depth: int = 12,
            num_heads: int = 12,
            mlp_ratio: float = 4.,
            qkv_bias: bool = True,
            drop_rate: float = 0.,
            pos_drop_rate: float = 0.,
            proj_drop_rate: float = 0.,
            attn_drop_rate: float = 0.,
            drop_path_rate: float = 0.,
            block_layers: Type[nn.Module] = LayerScaleBlock,
            block_layers_token: Type[nn.Module] = LayerScaleBlockClassAttn,
            patch_layer: Type[nn.Module] = PatchEmbed,
            norm_layer: Callable = partial(LayerNorm, eps=1e-6),
            act_layer: Type[nn.Module] = nn.GELU,
            attn_block: Type[nn.Module] = TalkingHeadAttn,
            mlp_block: Type[nn.Module] = Mlp,
            init_values: float = 1e-4,
            attn_block_token_only: Type[nn.Module] = ClassAttn,
            mlp_block_token_only: Type[nn.Module] = Mlp,
            depth_token_only: int = 2,
            mlp_ratio_token_only: float = 4.0,
    ):
        super().__init__()
        assert global_pool in ('', 'token', 'avg')

        self.num_classes = num_classes
        self.global_pool = global_pool
        self.num_features = self.head_hidden_size = self.embed_dim = embed_dim
        self.grad_checkpointing = False

        self.patch_embed = patch_layer(
            img_size=img_size,
            patch_size=patch_size,
            in_chans=in_chans,
            embed_dim=embed_dim,
        )
        num_patches = self.patch_embed.num_patches
        r = self.patch_embed.feat_ratio() if hasattr(self.patch_embed, 'feat_ratio') else patch_size

        self.cls_token = nn.Parameter(torch.zeros(1, 1, embed_dim))
        self.pos_embed = nn.Parameter(torch.zeros(1, num_patches, embed_dim))
        self.pos_drop = nn.Dropout(p=pos_drop_rate)

        shared = dict(
            dim=embed_dim,
            num_heads=num_heads,
            qkv_bias=qkv_bias,
            norm_layer=norm_layer,
            act_layer=act_layer,
            init_values=init_values,
        )
        self.blocks = nn.Sequential(*[
            block_layers(
                mlp_ratio=mlp_ratio,
                proj_drop=proj_drop_rate,
                attn_drop=attn_drop_rate,
                drop_path=drop_path_rate,
                attn_block=attn_block,
                mlp_block=mlp_block,
                **shared,
            ) for _ in range(depth)])
        self.feature_info = [
            dict(num_chs=embed_dim, reduction=r, module=f'blocks.{i}') for i in range(depth)]

        # the class-attention tail updates only the cls token
        self.blocks_token_only = nn.ModuleList([
            block_layers_token(
                mlp_ratio=mlp_ratio_token_only,
                attn_block=attn_block_token_only,
                mlp_block=mlp_block_token_only,
                **shared,
            ) for _ in range(depth_token_only)])

        self.norm = norm_layer(embed_dim)

        self.head_drop = nn.Dropout(drop_rate)
        self.head = nn.Linear(embed_dim, num_classes) if num_classes > 0 else nn.Identity()

        trunc_normal_(self.pos_embed, std=.02)
        trunc_normal_(self.cls_token, std=.02)
        self.apply(self._init_weights)

    def _init_weights(self, m: nn.Module):
        if isinstance(m, nn.Linear):
            trunc_normal_(m.weight, std=.02)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.constant_(m.bias, 0)
        elif isinstance(m, nn.LayerNorm):
            nn.init.constant_(m.bias, 0)
            nn.init.constant_(m.weight, 1.0)

    @torch.jit.ignore
    def no_weight_decay(self):
        return {'pos_embed', 'cls_token'}

    @torch.jit.ignore
    def set_grad_checkpointing(self, enable: bool = True):
        self.grad_checkpointing = enable

    @torch.jit.ignore
    def group_matcher(self, coarse: bool = False) -> Callable:
        def _matcher(name: str):
            if any([name.startswith(n) for n in ('cls_token', 'pos_embed', 'patch_embed')]):
                return 0
            elif name.startswith('blocks.'):
                return int(name.split('.')[1]) + 1
            elif name.startswith('blocks_token_only.'):
                # overlap token only blocks with last blocks
                to_offset = len(self.blocks) - len(self.blocks_token_only) + 1
                return int(name.split('.')[1]) + to_offset
            elif name.startswith('norm.'):
                return len(self.blocks)
            else:
                return float('inf')
        return _matcher

    @torch.jit.ignore
    def get_classifier(self) -> nn.Module:
        return self.head

    def reset_classifier(self, num_classes: int, global_pool: Optional[str] = None):
        self.num_classes = num_classes
        if global_pool is not None:
            assert global_pool in ('', 'token', 'avg')
            self.global_pool = global_pool
        self.head = nn.Linear(self.num_features, num_classes) if num_classes > 0 else nn.Identity()

    def forward_intermediates(
            self,
            x: torch.Tensor,
            indices: Optional[Union[int, List[int]]] = None,
            norm: bool = False,
            stop_early: bool = False,
            output_fmt: str = 'NCHW',
            intermediates_only: bool = False,
    ) -> Union[List[torch.Tensor], Tuple[torch.Tensor, List[torch.Tensor]]]:
        assert output_fmt in ('NCHW', 'NLC'), 'Output format must be one of NCHW or NLC.'
        take_indices, max_index = feature_take_indices(len(self.blocks), indices)
        B, _, height, width = x.shape

        x = self.pos_drop(self.patch_embed(x) + self.pos_embed)
        run_to = len(self.blocks) if (torch.jit.is_scripting() or not stop_early) else max_index + 1
        collected = []
        for i, blk in enumerate(self.blocks[:run_to]):
            x = blk(x)
            if i in take_indices:
                collected.append(self.norm(x) if norm else x)

        if output_fmt == 'NCHW':
            H, W = self.patch_embed.dyn_feat_size((height, width))
            collected = [
                t.reshape(B, H, W, -1).permute(0, 3, 1, 2).contiguous() for t in collected]
        if intermediates_only:
            return collected
        # cls-attention tail skipped for intermediate extraction
        return self.norm(x), collected

    def prune_intermediate_layers(
            self,
            indices: Union[int, List[int]] = 1,
            prune_norm: bool = False,
            prune_head: bool = True,
    ):
        take_indices, max_index = feature_take_indices(len(self.blocks), indices)
        self.blocks = self.blocks[:max_index + 1]
        if prune_norm:
            self.norm = nn.Identity()
        if prune_head:
            self.blocks_token_only = nn.ModuleList()
            self.reset_classifier(0, '')
        return take_indices

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        x = self.patch_embed(x)
        x = x + self.pos_embed
        x = self.pos_drop(x)
        if self.grad_checkpointing and not torch.jit.is_scripting():
            x = checkpoint_seq(self.blocks, x)
        else:
            x = self.blocks(x)
        cls_tokens = self.cls_token.expand(x.shape[0], -1, -1)
        for blk in self.blocks_token_only:
            cls_tokens = blk(x, cls_tokens)
        x = torch.cat((cls_tokens, x), dim=1)
        x = self.norm(x)
        return x

    def forward_head(self, x: torch.Tensor, pre_logits: bool = False) -> torch.Tensor:
        if self.global_pool:
            x = x[:, 1:].mean(dim=1) if self.global_pool == 'avg' else x[:, 0]
        x = self.head_drop(x)
        return x if pre_logits else self.head(x)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.forward_features(x)
        x = self.forward_head(x)
        return x


def checkpoint_filter_fn(state_dict, model=None):
    if 'model' in state_dict:
        state_dict = state_dict['model']
    checkpoint_no_module = {}
    for k, v in state_dict.items():
        checkpoint_no_module[k.replace('module.', '')] = v
    return checkpoint_no_module


def _create_cait(variant: str, pretrained: bool = False, **kwargs) -> Cait:
    out_indices = kwargs.pop('out_indices', 3)
    model = build_model_with_cfg(
        Cait,
        variant,
        pretrained,
        pretrained_filter_fn=checkpoint_filter_fn,
        feature_cfg=dict(out_indices=out_indices, feature_cls='getter'),
        **kwargs,
    )
    return model


def _cfg(url: str = '', **kwargs) -> Dict[str, Any]:
    return {
        'url': url,
        'num_classes': 1000, 'input_size': (3, 384, 384), 'pool_size': None,
        'crop_pct': 1.0, 'interpolation': 'bicubic', 'fixed_input_size': True,
        'mean': IMAGENET_DEFAULT_MEAN, 'std': IMAGENET_DEFAULT_STD,
        'first_conv': 'patch_embed.proj', 'classifier': 'head',
        **kwargs,
    }


default_cfgs = generate_default_cfgs({
    'cait_xxs24_224.fb_dist_in1k': _cfg(input_size=(3, 224, 224)),
    'cait_xxs24_384.fb_dist_in1k': _cfg(),
    'cait_xxs36_224.fb_dist_in1k': _cfg(input_size=(3, 224, 224)),
    'cait_xxs36_384.fb_dist_in1k': _cfg(),
    'cait_xs24_384.fb_dist_in1k': _cfg(),
    'cait_s24_224.fb_dist_in1k': _cfg(input_size=(3, 224, 224)),
    'cait_s24_384.fb_dist_in1k': _cfg(),
    'cait_s36_384.fb_dist_in1k': _cfg(),
    'cait_m36_384.fb_dist_in1k': _cfg(),
    'cait_m48_448.fb_dist_in1k': _cfg(input_size=(3, 448, 448)),
})


_VARIANTS = dict(
    cait_xxs24=dict(patch_size=16, embed_dim=192, depth=24, num_heads=4, init_values=1e-5),
    cait_xxs36=dict(patch_size=16, embed_dim=192, depth=36, num_heads=4, init_values=1e-5),
    cait_xs24=dict(patch_size=16, embed_dim=288, depth=24, num_heads=6, init_values=1e-5),
    cait_s24=dict(patch_size=16, embed_dim=384, depth=24, num_heads=8, init_values=1e-5),
    cait_s36=dict(patch_size=16, embed_dim=384, depth=36, num_heads=8, init_values=1e-6),
    cait_m36=dict(patch_size=16, embed_dim=768, depth=36, num_heads=16, init_values=1e-6),
    cait_m48=dict(patch_size=16, embed_dim=768, depth=48, num_heads=16, init_values=1e-6),
)


def _variant_args(name: str, kwargs: dict) -> dict:
    base = _VARIANTS[name.rsplit('_', 1)[0]]
    return dict(base, **kwargs)


@register_model
def cait_xxs24_224(pretrained=False, **kwargs) -> Cait:
    return _create_cait('cait_xxs24_224', pretrained=pretrained, **_variant_args('cait_xxs24_224', kwargs))


@register_model
def cait_xxs24_384(pretrained=False, **kwargs) -> Cait:
    return _create_cait('cait_xxs24_384', pretrained=pretrained, **_variant_args('cait_xxs24_384', kwargs))


@register_model
def cait_xxs36_224(pretrained=False, **kwargs) -> Cait:
    return _create_cait('cait_xxs36_224', pretrained=pretrained, **_variant_args('cait_xxs36_224', kwargs))


@register_model
def cait_xxs36_384(pretrained=False, **kwargs) -> Cait:
    return _create_cait('cait_xxs36_384', pretrained=pretrained, **_variant_args('cait_xxs36_384', kwargs))


@register_model
def cait_xs24_384(pretrained=False, **kwargs) -> Cait:
    return _create_cait('cait_xs24_384', pretrained=pretrained, **_variant_args('cait_xs24_384', kwargs))


@register_model
def cait_s24_224(pretrained=False, **kwargs) -> Cait:
    return _create_cait('cait_s24_224', pretrained=pretrained, **_variant_args('cait_s24_224', kwargs))


@register_model
def cait_s24_384(pretrained=False, **kwargs) -> Cait:
    return _create_cait('cait_s24_384', pretrained=pretrained, **_variant_args('cait_s24_384', kwargs))


@register_model
def cait_s36_384(pretrained=False, **kwargs) -> Cait:
    return _create_cait('cait_s36_384', pretrained=pretrained, **_variant_args('cait_s36_384', kwargs))


@register_model
def cait_m36_384(pretrained=False, **kwargs) -> Cait:
    return _create_cait('cait_m36_384', pretrained=pretrained, **_variant_args('cait_m36_384', kwargs))


@register_model
def cait_m48_448(pretrained=False, **kwargs) -> Cait:
    return _create_cait('cait_m48_448', pretrained=pretrained, **_variant_args('cait_m48_448', kwargs))
