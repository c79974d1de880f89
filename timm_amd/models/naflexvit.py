"""NaFlexVit — variable aspect/resolution ViT (NaFlex / NaViT / FlexiViT style).

Capability parity with reference `timm/models/naflexvit.py` (2,392 LoC):
`NaFlexVitCfg` (:59), `batch_patchify` (:163), `NaFlexEmbeds` (:339 — linear
patch-embed on pre-patchified input + pos-embed interpolation to coords),
`create_attention_mask` (:996 — additive -inf padding mask),
`global_pool_naflex` (:1065 — masked avg/max), `NaFlexVit` (:1137) consuming
`{patches[B,N,P*P*C], patch_coord[B,N,2], patch_valid[B,N]}` dicts
(forward :1906-1954).

MI355X mapping: the fused HIP attention kernel takes the additive
[B,1,N,N] padding mask directly; seq-len buckets are multiples of 8 so the
vectorized mask/softmax paths stay on the fast path.
"""
import math
from dataclasses import dataclass, field, fields, replace
from functools import partial
from typing import Any, Callable, Dict, List, Optional, Set, Tuple, Type, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..data.constants import IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD, OPENAI_CLIP_MEAN, OPENAI_CLIP_STD
from ..layers import (
    AttentionPoolLatent, LayerNorm, Mlp, PatchDropout, RmsNorm, RotaryEmbeddingCat,
    calculate_drop_path_rates, get_act_layer, get_norm_layer, to_2tuple, trunc_normal_,
)
from ._builder import build_model_with_cfg
from ._features import feature_take_indices
from ._manipulate import checkpoint
from ._registry import generate_default_cfgs, register_model
from .vision_transformer import Block

__all__ = ['NaFlexVit', 'NaFlexVitCfg', 'batch_patchify', 'create_attention_mask', 'global_pool_naflex']


@dataclass
class NaFlexVitCfg:
    """Configuration for NaFlexVit (reference `naflexvit.py:59`)."""
    patch_size: Union[int, Tuple[int, int]] = 16
    embed_dim: int = 768
    depth: int = 12
    num_heads: int = 12
    mlp_ratio: float = 4.
    qkv_bias: bool = True
    qk_norm: bool = False
    init_values: Optional[float] = None
    class_token: bool = False
    reg_tokens: int = 0
    pos_embed: str = 'learned'          # 'learned' | 'factorized' | 'rope' | 'learned_rope' | 'none'
    pos_embed_grid_size: Tuple[int, int] = (16, 16)
    pos_embed_interp_mode: str = 'bicubic'
    pos_embed_ar_preserving: bool = False  # interpolate to a square grid, then crop, so AR is kept
    rope_ref_feat_shape: Optional[Tuple[int, int]] = (16, 16)
    final_norm: bool = True
    fc_norm: Optional[bool] = None
    drop_rate: float = 0.
    pos_drop_rate: float = 0.
    patch_drop_rate: float = 0.
    proj_drop_rate: float = 0.
    attn_drop_rate: float = 0.
    drop_path_rate: float = 0.
    norm_layer: Optional[str] = None
    act_layer: Optional[str] = None
    embed_norm_layer: Optional[str] = None
    global_pool: str = 'map'            # 'map' | 'avg' | 'max' | 'token' | ''
    weight_init: str = ''


def batch_patchify(
        x: torch.Tensor,
        patch_size: Tuple[int, int],
        pad: bool = True,
        channels_last: bool = True,
) -> Tuple[torch.Tensor, Tuple[int, int]]:
    """[B,C,H,W] -> ([B, N, P*P*C], (nH, nW)) patchification (reference `:163`).

    channels_last=True gives P-P-C flat patches (NaFlex default); False gives
    C-P-P flat (HF / Gemma4 native layout).
    """
    B, C, H, W = x.shape
    ph, pw = patch_size
    if pad and (H % ph or W % pw):
        x = F.pad(x, (0, (pw - W % pw) % pw, 0, (ph - H % ph) % ph))
        B, C, H, W = x.shape
    nh, nw = H // ph, W // pw
    if channels_last:
        patches = x.view(B, C, nh, ph, nw, pw).permute(0, 2, 4, 3, 5, 1)  # (B, nh, nw, ph, pw, C)
    else:
        patches = x.view(B, C, nh, ph, nw, pw).permute(0, 2, 4, 1, 3, 5)  # (B, nh, nw, C, ph, pw)
    patches = patches.reshape(B, nh * nw, ph * pw * C)
    return patches, (nh, nw)


def create_attention_mask(
        patch_valid: torch.Tensor,
        num_prefix_tokens: int = 0,
        symmetric: bool = True,
        q_len: Optional[int] = None,
        dtype: torch.dtype = torch.float32,
) -> torch.Tensor:
    """Additive -inf padding mask [B,1,Q,N] from patch_valid [B,N] (reference `:996-1061`)."""
    patch_valid = patch_valid.bool()
    B, N = patch_valid.shape
    if num_prefix_tokens:
        prefix = patch_valid.new_ones(B, num_prefix_tokens)
        patch_valid = torch.cat([prefix, patch_valid], dim=1)
        N += num_prefix_tokens

    if symmetric:
        mask_bool = patch_valid.unsqueeze(-1) & patch_valid.unsqueeze(1)  # [B,N,N]
        mask_bool = mask_bool.unsqueeze(1)  # [B,1,N,N]
    else:
        q_len = q_len or N
        mask_bool = patch_valid[:, None, None, :].expand(B, 1, q_len, N)

    mask_float = torch.zeros_like(mask_bool, dtype=dtype)
    mask_float.masked_fill_(~mask_bool, torch.finfo(dtype).min)
    return mask_float


def global_pool_naflex(
        x: torch.Tensor,
        patch_valid: Optional[torch.Tensor] = None,
        pool_type: str = 'avg',
        num_prefix_tokens: int = 1,
) -> torch.Tensor:
    """Masked global pooling over valid patches (reference `:1065`)."""
    if patch_valid is None or pool_type not in ('avg', 'avgmax', 'max'):
        # fall back to unmasked pooling
        if pool_type == 'token':
            return x[:, 0]
        x = x[:, num_prefix_tokens:]
        if pool_type == 'avg':
            return x.mean(dim=1)
        if pool_type == 'max':
            return x.amax(dim=1)
        if pool_type == 'avgmax':
            return 0.5 * (x.mean(dim=1) + x.amax(dim=1))
        return x

    if num_prefix_tokens:
        x = x[:, num_prefix_tokens:]
    if x.is_cuda:
        # fused mask+reduce kernel (csrc/data_ops.hip); bwd from count/argmax
        from .. import ops
        return ops.masked_global_pool(x, patch_valid, pool_type)
    patch_valid = patch_valid.to(x.dtype)
    denom = patch_valid.sum(dim=1, keepdim=True).clamp(min=1)
    if pool_type == 'avg':
        return (x * patch_valid.unsqueeze(-1)).sum(dim=1) / denom
    masked = x.masked_fill(~patch_valid.bool().unsqueeze(-1), torch.finfo(x.dtype).min)
    if pool_type == 'max':
        return masked.amax(dim=1)
    # avgmax
    avg = (x * patch_valid.unsqueeze(-1)).sum(dim=1) / denom
    return 0.5 * (avg + masked.amax(dim=1))


class NaFlexEmbeds(nn.Module):
    """Linear patch embed on pre-patchified input + pos embed resampled to
    patch coords (reference `naflexvit.py:339`)."""

    def __init__(
            self,
            patch_size: int = 16,
            in_chans: int = 3,
            embed_dim: int = 768,
            pos_embed: str = 'learned',
            pos_embed_grid_size: Tuple[int, int] = (16, 16),
            pos_embed_interp_mode: str = 'bicubic',
            pos_embed_ar_preserving: bool = False,
            proj_norm_layer: Optional[Type[nn.Module]] = None,
            pos_drop_rate: float = 0.,
            class_token: bool = False,
            reg_tokens: int = 0,
            bias: bool = True,
            input_norm_layer: Optional[Type[nn.Module]] = None,
    ):
        super().__init__()
        self.patch_size = to_2tuple(patch_size)
        self.in_chans = in_chans
        self.embed_dim = embed_dim
        self.pos_embed_type = pos_embed
        self.pos_embed_interp_mode = pos_embed_interp_mode
        self.pos_embed_ar_preserving = pos_embed_ar_preserving
        self.pos_embed_grid_size = pos_embed_grid_size
        self.num_prefix_tokens = (1 if class_token else 0) + reg_tokens

        self.cls_token = nn.Parameter(torch.zeros(1, 1, embed_dim)) if class_token else None
        self.reg_token = nn.Parameter(torch.zeros(1, reg_tokens, embed_dim)) if reg_tokens else None

        patch_dim = self.patch_size[0] * self.patch_size[1] * in_chans
        self.norm_input = input_norm_layer(patch_dim) if input_norm_layer else None
        self.proj = nn.Linear(patch_dim, embed_dim, bias=bias)
        self.norm_proj = proj_norm_layer(embed_dim) if proj_norm_layer else nn.Identity()

        if pos_embed in ('learned', 'learned_rope'):
            self.pos_embed = nn.Parameter(
                torch.randn(1, pos_embed_grid_size[0], pos_embed_grid_size[1], embed_dim) * .02)
        else:
            self.pos_embed = None
        self.pos_drop = nn.Dropout(p=pos_drop_rate)

    def feat_ratio(self, as_scalar=True):
        if as_scalar:
            return max(self.patch_size)
        return self.patch_size

    def dyn_feat_size(self, img_size: Tuple[int, int]) -> Tuple[int, int]:
        return img_size[0] // self.patch_size[0], img_size[1] // self.patch_size[1]

    def _interp_pos_embed(self, patch_coord: torch.Tensor, grid_size: Tuple[int, int]) -> torch.Tensor:
        """Interpolate the learned pos-embed grid to each sample's patch grid,
        then gather per patch coord (reference `:613-884`)."""
        B, N = patch_coord.shape[0], patch_coord.shape[1]
        # resize pos embed grid to the max grid of the batch then index by coords
        pe = self.pos_embed.permute(0, 3, 1, 2)  # [1, C, gh, gw]
        gh, gw = grid_size
        if (gh, gw) != tuple(pe.shape[-2:]):
            # ar-preserving: scale the square table uniformly to cover the larger
            # side, then crop — both axes keep the same stretch factor
            interp_size = (max(gh, gw),) * 2 if self.pos_embed_ar_preserving else (gh, gw)
            pe = F.interpolate(
                pe.float(), size=interp_size, mode=self.pos_embed_interp_mode,
                antialias=True, align_corners=False).to(pe.dtype)
            pe = pe[:, :, :gh, :gw]
        pe = pe.permute(0, 2, 3, 1)  # [1, gh, gw, C]
        flat = pe.reshape(1, gh * gw, self.embed_dim).expand(B, -1, -1)
        idx = (patch_coord[..., 0].clamp(max=gh - 1) * gw + patch_coord[..., 1].clamp(max=gw - 1)).long()
        return flat.gather(1, idx.unsqueeze(-1).expand(-1, -1, self.embed_dim))

    def forward(
            self,
            patches: torch.Tensor,
            patch_coord: Optional[torch.Tensor] = None,
            patch_valid: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        if self.norm_input is not None:
            patches = self.norm_input(patches)
        x = self.proj(patches)
        x = self.norm_proj(x)

        if self.pos_embed is not None:
            if patch_coord is not None:
                gh = int(patch_coord[..., 0].amax().item()) + 1
                gw = int(patch_coord[..., 1].amax().item()) + 1
                pos = self._interp_pos_embed(patch_coord, (gh, gw))
            else:
                # fixed grid (plain image path); assume square
                n = x.shape[1]
                g = int(math.sqrt(n))
                coords = torch.stack(torch.meshgrid(
                    torch.arange(g, device=x.device), torch.arange(g, device=x.device),
                    indexing='ij'), -1).reshape(1, n, 2).expand(x.shape[0], -1, -1)
                pos = self._interp_pos_embed(coords, (g, g))
            x = x + pos.to(x.dtype)

        to_cat = []
        if self.cls_token is not None:
            to_cat.append(self.cls_token.expand(x.shape[0], -1, -1))
        if self.reg_token is not None:
            to_cat.append(self.reg_token.expand(x.shape[0], -1, -1))
        if to_cat:
            x = torch.cat(to_cat + [x], dim=1)

        return self.pos_drop(x)


class NaFlexVit(nn.Module):
    """Vision Transformer supporting NaFlex variable-resolution input
    (reference `naflexvit.py:1137`).

    Accepts either a plain [B,C,H,W] image tensor or a dict with
    `patches` [B,N,P*P*C], `patch_coord` [B,N,2] (row, col), and
    `patch_valid` [B,N] bool.
    """

    def __init__(
            self,
            cfg: Optional[NaFlexVitCfg] = None,
            in_chans: int = 3,
            num_classes: int = 1000,
            img_size: Optional[int] = None,
            **kwargs,
    ):
        super().__init__()
        cfg = cfg or NaFlexVitCfg()
        cfg_fields = {f.name for f in fields(NaFlexVitCfg)}
        overlay = {k: v for k, v in kwargs.items() if k in cfg_fields}
        if overlay:
            cfg = replace(cfg, **overlay)
        self.cfg = cfg

        norm_layer = get_norm_layer(cfg.norm_layer) or partial(LayerNorm, eps=1e-6)
        embed_norm_layer = get_norm_layer(cfg.embed_norm_layer)
        act_layer = get_act_layer(cfg.act_layer) or nn.GELU

        self.num_classes = num_classes
        self.global_pool = cfg.global_pool
        self.num_features = self.head_hidden_size = self.embed_dim = cfg.embed_dim
        self.num_prefix_tokens = (1 if cfg.class_token else 0) + cfg.reg_tokens
        self.grad_checkpointing = False

        self.embeds = NaFlexEmbeds(
            patch_size=cfg.patch_size,
            in_chans=in_chans,
            embed_dim=cfg.embed_dim,
            pos_embed=cfg.pos_embed,
            pos_embed_grid_size=tuple(cfg.pos_embed_grid_size),
            pos_embed_ar_preserving=cfg.pos_embed_ar_preserving,
            pos_embed_interp_mode=cfg.pos_embed_interp_mode,
            proj_norm_layer=embed_norm_layer,
            pos_drop_rate=cfg.pos_drop_rate,
            class_token=cfg.class_token,
            reg_tokens=cfg.reg_tokens,
        )
        # alias for first_conv compat naming / checkpoint mapping
        self.patch_embed = self.embeds

        if 'rope' in cfg.pos_embed:
            self.rope = RotaryEmbeddingCat(
                cfg.embed_dim // cfg.num_heads,
                in_pixels=False,
                feat_shape=None,
                ref_feat_shape=cfg.rope_ref_feat_shape,
            )
        else:
            self.rope = None

        dpr = calculate_drop_path_rates(cfg.drop_path_rate, cfg.depth)
        if self.rope is not None:
            # rope needs a block whose attention applies the rotary embed past
            # the prefix tokens (reference `naflexvit.py:291-335` uses the EVA
            # block family for all rope attn types)
            from .eva import EvaBlock
            self.blocks = nn.ModuleList([
                EvaBlock(
                    dim=cfg.embed_dim,
                    num_heads=cfg.num_heads,
                    mlp_ratio=cfg.mlp_ratio,
                    qkv_bias=cfg.qkv_bias,
                    attn_type='rope',
                    num_prefix_tokens=self.num_prefix_tokens,
                    init_values=cfg.init_values,
                    proj_drop=cfg.proj_drop_rate,
                    attn_drop=cfg.attn_drop_rate,
                    drop_path=dpr[i],
                    norm_layer=norm_layer,
                    act_layer=act_layer,
                )
                for i in range(cfg.depth)])
        else:
            self.blocks = nn.ModuleList([
                Block(
                    dim=cfg.embed_dim,
                    num_heads=cfg.num_heads,
                    mlp_ratio=cfg.mlp_ratio,
                    qkv_bias=cfg.qkv_bias,
                    qk_norm=cfg.qk_norm,
                    init_values=cfg.init_values,
                    proj_drop=cfg.proj_drop_rate,
                    attn_drop=cfg.attn_drop_rate,
                    drop_path=dpr[i],
                    norm_layer=norm_layer,
                    act_layer=act_layer,
                )
                for i in range(cfg.depth)])
        self.feature_info = [
            dict(module=f'blocks.{i}', num_chs=cfg.embed_dim, reduction=cfg.patch_size)
            for i in range(cfg.depth)]

        use_fc_norm = cfg.global_pool in ('avg', 'avgmax', 'max') if cfg.fc_norm is None else cfg.fc_norm
        self.norm = norm_layer(cfg.embed_dim) if cfg.final_norm and not use_fc_norm else nn.Identity()

        if cfg.global_pool == 'map':
            self.attn_pool = AttentionPoolLatent(
                self.embed_dim,
                num_heads=cfg.num_heads,
                mlp_ratio=cfg.mlp_ratio,
                norm_layer=norm_layer,
                act_layer=act_layer,
            )
        else:
            self.attn_pool = None
        self.fc_norm = norm_layer(cfg.embed_dim) if cfg.final_norm and use_fc_norm else nn.Identity()
        self.head_drop = nn.Dropout(cfg.drop_rate)
        self.head = nn.Linear(self.embed_dim, num_classes) if num_classes > 0 else nn.Identity()

        if cfg.weight_init != 'skip':
            self.init_weights()

    def init_weights(self):
        if self.embeds.pos_embed is not None:
            trunc_normal_(self.embeds.pos_embed, std=.02)
        if self.embeds.cls_token is not None:
            nn.init.normal_(self.embeds.cls_token, std=1e-6)
        from ..layers import init_weight_vit
        from ._manipulate import named_apply
        named_apply(init_weight_vit, self)

    @torch.jit.ignore
    def no_weight_decay(self) -> Set:
        return {'embeds.pos_embed', 'embeds.cls_token', 'embeds.reg_token'}

    @torch.jit.ignore
    def group_matcher(self, coarse: bool = False) -> Dict:
        return dict(
            stem=r'^embeds|^patch_embed',
            blocks=[(r'^blocks\.(\d+)', None), (r'^norm', (99999,))]
        )

    @torch.jit.ignore
    def set_grad_checkpointing(self, enable: bool = True) -> None:
        self.grad_checkpointing = enable

    @torch.jit.ignore
    def get_classifier(self) -> nn.Module:
        return self.head

    def reset_classifier(self, num_classes: int, global_pool: Optional[str] = None):
        self.num_classes = num_classes
        if global_pool is not None:
            if global_pool != 'map' and self.attn_pool is not None:
                self.attn_pool = None
            self.global_pool = global_pool
        self.head = nn.Linear(self.embed_dim, num_classes) if num_classes > 0 else nn.Identity()

    def _unpack(self, x):
        if isinstance(x, dict):
            patches = x['patches']
            patch_coord = x.get('patch_coord', None)
            patch_valid = x.get('patch_valid', None)
            grid = None
        else:
            patches, grid = batch_patchify(x, self.embeds.patch_size)
            patch_coord = None
            patch_valid = None
        return patches, patch_coord, patch_valid, grid

    def forward_features(self, x) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
        patches, patch_coord, patch_valid, _ = self._unpack(x)
        h = self.embeds(patches, patch_coord=patch_coord, patch_valid=patch_valid)

        attn_mask = None
        if patch_valid is not None:
            attn_mask = create_attention_mask(
                patch_valid, num_prefix_tokens=self.num_prefix_tokens, dtype=torch.float32)

        rope_embed = None
        if self.rope is not None:
            if patch_coord is not None:
                shapes = [(int(patch_coord[i, :, 0].amax()) + 1, int(patch_coord[i, :, 1].amax()) + 1)
                          for i in range(patch_coord.shape[0])]
                rope_embed = self.rope.get_batch_embeds(shapes, seq_len=patches.shape[1])
            else:
                n = patches.shape[1]
                g = int(math.sqrt(n))
                rope_embed = self.rope.get_embed((g, g))

        for blk in self.blocks:
            blk_kwargs = dict(attn_mask=attn_mask)
            if rope_embed is not None:
                blk_kwargs['rope'] = rope_embed
            if self.grad_checkpointing and not torch.jit.is_scripting():
                h = checkpoint(blk, h, **blk_kwargs)
            else:
                h = blk(h, **blk_kwargs)
        h = self.norm(h)
        return h, patch_valid

    def forward_head(self, x: torch.Tensor, patch_valid: Optional[torch.Tensor] = None,
                     pre_logits: bool = False) -> torch.Tensor:
        if self.attn_pool is not None:
            attn_mask = None
            if patch_valid is not None:
                attn_mask = create_attention_mask(
                    patch_valid, num_prefix_tokens=self.num_prefix_tokens,
                    symmetric=False, q_len=self.attn_pool.latent_len, dtype=torch.float32)
            x = self.attn_pool(x, attn_mask=attn_mask)
        else:
            x = global_pool_naflex(
                x, patch_valid, pool_type=self.global_pool,
                num_prefix_tokens=self.num_prefix_tokens)
        x = self.fc_norm(x)
        x = self.head_drop(x)
        return x if pre_logits else self.head(x)

    def forward_intermediates(
            self,
            x,
            indices: Optional[Union[int, List[int]]] = None,
            norm: bool = False,
            stop_early: bool = False,
            output_fmt: str = 'NLC',
            intermediates_only: bool = False,
    ):
        assert output_fmt in ('NLC',), 'NaFlexVit forward_intermediates is NLC only'
        take_indices, max_index = feature_take_indices(len(self.blocks), indices)
        patches, patch_coord, patch_valid, _ = self._unpack(x)
        h = self.embeds(patches, patch_coord=patch_coord, patch_valid=patch_valid)
        attn_mask = None
        if patch_valid is not None:
            attn_mask = create_attention_mask(
                patch_valid, num_prefix_tokens=self.num_prefix_tokens, dtype=torch.float32)
        intermediates = []
        blocks = self.blocks if not stop_early else self.blocks[:max_index + 1]
        for i, blk in enumerate(blocks):
            h = blk(h, attn_mask=attn_mask)
            if i in take_indices:
                intermediates.append(self.norm(h) if norm else h)
        if intermediates_only:
            return intermediates
        h = self.norm(h)
        return h, intermediates

    def forward(self, x) -> torch.Tensor:
        h, patch_valid = self.forward_features(x)
        return self.forward_head(h, patch_valid)


def checkpoint_filter_fn(state_dict, model):
    """Remap plain-ViT checkpoints onto NaFlexVit naming."""
    state_dict = state_dict.get('model', state_dict)
    out = {}
    for k, v in state_dict.items():
        if k == 'pos_embed' and v.ndim == 3:
            # [1, N, C] -> [1, gh, gw, C]
            n = v.shape[1]
            g = int(math.sqrt(n))
            if g * g == n:
                v = v.reshape(1, g, g, v.shape[-1])
            k = 'embeds.pos_embed'
        elif k == 'patch_embed.proj.weight' and v.ndim == 4:
            # conv [D, C, ph, pw] -> linear [D, ph*pw*C] in P-P-C order
            D, C, ph, pw = v.shape
            v = v.permute(0, 2, 3, 1).reshape(D, ph * pw * C)
            k = 'embeds.proj.weight'
        elif k == 'patch_embed.proj.bias':
            k = 'embeds.proj.bias'
        elif k == 'cls_token':
            k = 'embeds.cls_token'
        elif k == 'reg_token':
            k = 'embeds.reg_token'
        out[k] = v
    return out


def _cfg(url: str = '', **kwargs):
    return {
        'url': url,
        'num_classes': 1000, 'input_size': (3, 384, 384), 'pool_size': None,
        'crop_pct': 1.0, 'interpolation': 'bicubic', 'fixed_input_size': False,
        'mean': IMAGENET_DEFAULT_MEAN, 'std': IMAGENET_DEFAULT_STD,
        'first_conv': 'embeds.proj', 'classifier': 'head',
        **kwargs,
    }


default_cfgs = generate_default_cfgs({
    'naflexvit_base_patch16_gap.e300_s576_in1k': _cfg(
        hf_hub_id='timm/',
    ),
    'naflexvit_base_patch16_par_gap.e300_s576_in1k': _cfg(
        hf_hub_id='timm/',
    ),
    'naflexvit_base_patch16_parfac_gap.e300_s576_in1k': _cfg(
        hf_hub_id='timm/',
    ),
    'naflexvit_base_patch16_map.untrained': _cfg(),
    'naflexvit_so150m2_patch16_reg1_gap.untrained': _cfg(),
    'naflexvit_so150m2_patch16_reg1_map.untrained': _cfg(),

    # SigLIP-2 NaFlex vit encoder weights
    'naflexvit_base_patch16_siglip.v2_webli': _cfg(
        hf_hub_id='timm/',
        num_classes=0),
    'naflexvit_so400m_patch16_siglip.v2_webli': _cfg(
        hf_hub_id='timm/',
        num_classes=0),
})


def _create_naflexvit(variant: str, pretrained: bool = False, **kwargs) -> NaFlexVit:
    out_indices = kwargs.pop('out_indices', 3)
    model = build_model_with_cfg(
        NaFlexVit, variant, pretrained,
        pretrained_filter_fn=checkpoint_filter_fn,
        feature_cfg=dict(out_indices=out_indices, feature_cls='getter'),
        kwargs_filter=('img_size',),
        **kwargs,
    )
    return model


@register_model
def naflexvit_base_patch16_gap(pretrained: bool = False, **kwargs) -> NaFlexVit:
    """NaFlexVit-Base w/ global average pooling — BASELINE config #5."""
    cfg = NaFlexVitCfg(
        patch_size=16, embed_dim=768, depth=12, num_heads=12,
        global_pool='avg', reg_tokens=4, fc_norm=True,
    )
    return _create_naflexvit('naflexvit_base_patch16_gap', pretrained=pretrained, cfg=cfg, **kwargs)


@register_model
def naflexvit_base_patch16_map(pretrained: bool = False, **kwargs) -> NaFlexVit:
    cfg = NaFlexVitCfg(
        patch_size=16, embed_dim=768, depth=12, num_heads=12,
        global_pool='map',
    )
    return _create_naflexvit('naflexvit_base_patch16_map', pretrained=pretrained, cfg=cfg, **kwargs)


@register_model
def naflexvit_base_patch16_siglip(pretrained: bool = False, **kwargs) -> NaFlexVit:
    cfg = NaFlexVitCfg(
        patch_size=16, embed_dim=768, depth=12, num_heads=12,
        global_pool='map', class_token=False,
    )
    return _create_naflexvit('naflexvit_base_patch16_siglip', pretrained=pretrained, cfg=cfg, **kwargs)


@register_model
def naflexvit_so150m2_patch16_reg1_gap(pretrained: bool = False, **kwargs) -> NaFlexVit:
    cfg = NaFlexVitCfg(
        patch_size=16, embed_dim=832, depth=21, num_heads=13, mlp_ratio=34 / 8,
        init_values=1e-5, qkv_bias=False,
        global_pool='avg', reg_tokens=1, fc_norm=True,
    )
    return _create_naflexvit('naflexvit_so150m2_patch16_reg1_gap', pretrained=pretrained, cfg=cfg, **kwargs)


@register_model
def naflexvit_base_patch16_par_gap(pretrained: bool = False, **kwargs) -> NaFlexVit:
    """ViT-Base with NaFlex functionality, aspect preserving pos embed, global average pooling."""
    cfg = NaFlexVitCfg(
        patch_size=16,
        embed_dim=768,
        depth=12,
        num_heads=12,
        init_values=1e-5,
        pos_embed_ar_preserving=True,
        global_pool='avg',
        reg_tokens=4,
        fc_norm=True,
    )
    model = _create_naflexvit('naflexvit_base_patch16_par_gap', pretrained=pretrained, cfg=cfg, **kwargs)
    return model


@register_model
def naflexvit_base_patch16_parfac_gap(pretrained: bool = False, **kwargs) -> NaFlexVit:
    """ViT-Base with NaFlex functionality, aspect preserving & factorized pos embed, global average pooling."""
    cfg = NaFlexVitCfg(
        patch_size=16,
        embed_dim=768,
        depth=12,
        num_heads=12,
        init_values=1e-5,
        pos_embed_ar_preserving=True,
        pos_embed='factorized',
        global_pool='avg',
        reg_tokens=4,
        fc_norm=True,
    )
    model = _create_naflexvit('naflexvit_base_patch16_parfac_gap', pretrained=pretrained, cfg=cfg, **kwargs)
    return model


@register_model
def naflexvit_so150m2_patch16_reg1_map(pretrained: bool = False, **kwargs) -> NaFlexVit:
    """ViT-SO150M2 with NaFlex functionality for variable aspect ratios and resolutions. This model supports: 1. Variable aspect ratios and resolutions via patch coordinates 2. Position embedding interpolation for arbitrary grid sizes 3. Explicit patch coordinates and valid token masking"""
    cfg = NaFlexVitCfg(
        patch_size=16,
        embed_dim=832,
        depth=21,
        num_heads=13,
        mlp_ratio=34/13,
        init_values=1e-5,
        qkv_bias=False,
        reg_tokens=1,
        global_pool='map',
    )
    model = _create_naflexvit('naflexvit_so150m2_patch16_reg1_map', pretrained=pretrained, cfg=cfg, **kwargs)
    return model


@register_model
def naflexvit_so400m_patch16_siglip(pretrained: bool = False, **kwargs) -> NaFlexVit:
    """ViT-SO400M with NaFlex functionality for variable aspect ratios and resolutions."""
    cfg = NaFlexVitCfg(
        patch_size=16,
        embed_dim=1152,
        depth=27,
        num_heads=16,
        mlp_ratio=3.7362,
        act_layer='gelu_tanh',
        global_pool='map',
    )
    model = _create_naflexvit('naflexvit_so400m_patch16_siglip', pretrained=pretrained, cfg=cfg, **kwargs)
    return model

