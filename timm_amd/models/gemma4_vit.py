"""Gemma4 vision tower — ViT with 2D RoPE, gated MLP, QKV RMSNorm, sandwich norms.

Capability parity with reference `timm/models/gemma4_vit.py`:
`Gemma4ClippableLinear` (:32, clamp-bracketed projections in the E4B tower),
2D rotary embedding (:122), linear patch embed + one-hot 2D position table
(:202) accepting raw/NaFlex inputs, QKV-normalized attention at scale 1.0
(:348), gated GELU-tanh MLP (:450), 4-norm sandwich blocks (:478), the
spatial k×k soft-token pooler (:551), the headless encoder (:618) and the
classifier wrapper (:996).

RMSNorms and the attention core route through our fused HIP kernels.
"""
import math
from functools import partial
from typing import Any, Callable, Dict, List, Optional, Set, Tuple, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from ..layers import DropPath, RmsNorm, to_2tuple, use_fused_attn
from ..layers.weight_init import trunc_normal_tf_
from ._builder import build_model_with_cfg
from ._features import feature_take_indices
from ._manipulate import checkpoint, named_apply
from ._registry import generate_default_cfgs, register_model
from .naflexvit import batch_patchify

__all__ = ['Gemma4VitEncoder', 'Gemma4VitClassifier']


class Gemma4ClippableLinear(nn.Module):
    """Bias-free linear with optional input/output clamping (E4B tower)."""

    def __init__(self, in_features: int, out_features: int, use_clipped: bool = False):
        super().__init__()
        self.use_clipped = use_clipped
        self.linear = nn.Linear(in_features, out_features, bias=False)
        if use_clipped:
            self.register_buffer('input_min', torch.empty(()))
            self.register_buffer('input_max', torch.empty(()))
            self.register_buffer('output_min', torch.empty(()))
            self.register_buffer('output_max', torch.empty(()))
            self.reset_parameters()

    def reset_parameters(self) -> None:
        # clamp buffers default to ±inf no-op; pretrained checkpoints overwrite
        if self.use_clipped:
            self.input_min.fill_(-float('inf'))
            self.input_max.fill_(float('inf'))
            self.output_min.fill_(-float('inf'))
            self.output_max.fill_(float('inf'))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.use_clipped:
            x = torch.clamp(x, self.input_min, self.input_max)
        x = self.linear(x)
        if self.use_clipped:
            x = torch.clamp(x, self.output_min, self.output_max)
        return x


def rotate_half(x: torch.Tensor) -> torch.Tensor:
    x1 = x[..., : x.shape[-1] // 2]
    x2 = x[..., x.shape[-1] // 2:]
    return torch.cat((-x2, x1), dim=-1)


def apply_rotary_pos_emb(x, cos, sin, unsqueeze_dim: int = 2):
    cos = cos.unsqueeze(unsqueeze_dim)
    sin = sin.unsqueeze(unsqueeze_dim)
    return (x * cos) + (rotate_half(x) * sin)


def apply_multidimensional_rope(x, cos, sin, ndim: int = 2, unsqueeze_dim: int = 2):
    """Split head_dim into ndim parts, rope each with its own cos/sin slice."""
    n = x.shape[-1]
    per_dim = 2 * (n // (2 * ndim))
    split = [per_dim] * ndim
    xs = torch.split(x, split, dim=-1)
    cs = torch.split(cos, split, dim=-1)
    ss = torch.split(sin, split, dim=-1)
    return torch.cat(
        [apply_rotary_pos_emb(xs[k], cs[k], ss[k], unsqueeze_dim=unsqueeze_dim) for k in range(ndim)], dim=-1)


class Gemma4RotaryEmbedding2D(nn.Module):
    """Per-axis rotary embedding with theta=100 over half the head dim each."""

    def __init__(self, head_dim: int, rope_theta: float = 100.0):
        super().__init__()
        self.head_dim = head_dim
        self.rope_theta = rope_theta
        num_freqs = (head_dim // 2) // 2
        self.register_buffer('inv_freq', torch.empty(num_freqs, dtype=torch.float), persistent=False)
        self._init_buffers()

    def _init_buffers(self) -> None:
        spatial_dim = self.head_dim // 2
        inv_freq = 1.0 / (
            self.rope_theta ** (torch.arange(0, spatial_dim, 2, dtype=torch.float, device=self.inv_freq.device) / spatial_dim))
        self.inv_freq.copy_(inv_freq)

    def reset_parameters(self) -> None:
        self._init_buffers()

    def forward(self, x: torch.Tensor, position_ids: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """position_ids: (B, N, 2) internal (x, y) coords → cos/sin (B, N, head_dim)."""
        with torch.no_grad():
            inv_freq = self.inv_freq[None, :, None].float().expand(position_ids.shape[0], -1, 1).to(x.device)
            all_cos: List[torch.Tensor] = []
            all_sin: List[torch.Tensor] = []
            for i in range(2):
                pos = position_ids[:, :, i][:, None, :].float()  # (B, 1, N)
                freqs = (inv_freq @ pos).transpose(1, 2)         # (B, N, sd//2)
                emb = torch.cat((freqs, freqs), dim=-1)
                all_cos.append(emb.cos())
                all_sin.append(emb.sin())
            cos = torch.cat(all_cos, dim=-1).to(dtype=x.dtype)
            sin = torch.cat(all_sin, dim=-1).to(dtype=x.dtype)
        return cos, sin


class Gemma4PatchEmbed(nn.Module):
    """Linear patch projection + 2D one-hot position table.

    Accepts raw (B,C,H,W), NaFlex flat (B,N,P*P*C), unflattened
    (B,N,Ph,Pw,C), or a dict with patches/patch_coord/patch_valid.
    """

    def __init__(
            self,
            patch_size: Union[int, Tuple[int, int]] = 16,
            in_chans: int = 3,
            embed_dim: int = 768,
            position_embedding_size: int = 10240,
    ):
        super().__init__()
        self.patch_size = to_2tuple(patch_size)
        self.embed_dim = embed_dim
        self.position_embedding_size = position_embedding_size
        ph, pw = self.patch_size
        self.input_proj = nn.Linear(in_chans * ph * pw, embed_dim, bias=False)
        self.position_embedding_table = nn.Parameter(torch.empty(2, position_embedding_size, embed_dim))
        self.reset_parameters()

    def reset_parameters(self) -> None:
        trunc_normal_tf_(self.position_embedding_table, std=0.02)

    def _default_patch_coord(self, batch_size: int, pH: int, pW: int, device):
        ys = torch.arange(pH, device=device)
        xs = torch.arange(pW, device=device)
        gy, gx = torch.meshgrid(ys, xs, indexing='ij')
        coord = torch.stack([gy.flatten(), gx.flatten()], dim=-1).unsqueeze(0).expand(batch_size, -1, -1)
        valid = torch.ones(batch_size, pH * pW, dtype=torch.bool, device=device)
        return coord, valid

    def _position_embeddings(self, position_ids: torch.Tensor, padding_positions: torch.Tensor) -> torch.Tensor:
        clamped = position_ids.clamp(min=0)
        one_hot = F.one_hot(clamped, num_classes=self.position_embedding_size)
        one_hot = one_hot.permute(0, 2, 1, 3).to(self.position_embedding_table)
        emb = (one_hot @ self.position_embedding_table).sum(dim=1)  # (B, N, D)
        return torch.where(padding_positions.unsqueeze(-1), 0.0, emb)

    def forward(
            self,
            x: Union[torch.Tensor, Dict[str, torch.Tensor]],
            patch_coord: Optional[torch.Tensor] = None,
            patch_valid: Optional[torch.Tensor] = None,
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        if isinstance(x, dict):
            patch_coord = x.get('patch_coord', patch_coord)
            patch_valid = x.get('patch_valid', patch_valid)
            x = x['patches']

        ph, pw = self.patch_size
        if x.ndim == 4:
            B, _, H, W = x.shape
            if patch_coord is None:
                patch_coord, patch_valid = self._default_patch_coord(B, H // ph, W // pw, x.device)
            x, _ = batch_patchify(x, (ph, pw), pad=False, channels_last=False)  # C-Ph-Pw flat
        elif x.ndim == 5:
            # (B, N, Ph, Pw, C) -> C-Ph-Pw flat
            x = x.permute(0, 1, 4, 2, 3).reshape(x.shape[0], x.shape[1], -1)
        elif x.ndim == 3:
            # NaFlex P-P-C flat -> C-Ph-Pw flat
            B, N, PPC = x.shape
            C = PPC // (ph * pw)
            x = x.view(B, N, ph, pw, C).permute(0, 1, 4, 2, 3).reshape(B, N, PPC)
        else:
            raise ValueError(f'Expected input ndim in (3, 4, 5); got {x.ndim}.')

        if patch_coord is None:
            raise ValueError('patch_coord is required for pre-patchified input.')
        if patch_valid is None:
            sentinel = (patch_coord == -1).all(dim=-1)
            if sentinel.any():
                patch_valid = ~sentinel
            else:
                patch_valid = torch.ones(patch_coord.shape[:2], dtype=torch.bool, device=patch_coord.device)

        # gemma4 native [0,1] -> [-1,1] pixel scaling
        x = 2 * (x - 0.5)
        x = self.input_proj(x.to(self.input_proj.weight.dtype))

        position_ids = patch_coord.flip(dims=(-1,))  # external (y,x) -> internal (x,y)
        padding_positions = ~patch_valid
        x = x + self._position_embeddings(position_ids, padding_positions)
        return x, position_ids, padding_positions


class Gemma4Attention(nn.Module):
    """Separate Q/K/V projections, RMSNorm on each head, 2D RoPE, scale=1."""
    fused_attn: torch.jit.Final[bool]

    def __init__(
            self,
            dim: int,
            num_heads: int = 12,
            head_dim: int = 64,
            num_kv_heads: Optional[int] = None,
            attn_drop: float = 0.0,
            proj_drop: float = 0.0,
            norm_eps: float = 1e-6,
            use_clipped_linears: bool = False,
    ):
        super().__init__()
        self.num_heads = num_heads
        self.head_dim = head_dim
        self.num_kv_heads = num_kv_heads or num_heads
        self.num_kv_groups = num_heads // self.num_kv_heads
        self.fused_attn = use_fused_attn()

        self.q_proj = Gemma4ClippableLinear(dim, num_heads * head_dim, use_clipped=use_clipped_linears)
        self.k_proj = Gemma4ClippableLinear(dim, self.num_kv_heads * head_dim, use_clipped=use_clipped_linears)
        self.v_proj = Gemma4ClippableLinear(dim, self.num_kv_heads * head_dim, use_clipped=use_clipped_linears)
        self.o_proj = Gemma4ClippableLinear(num_heads * head_dim, dim, use_clipped=use_clipped_linears)

        self.q_norm = RmsNorm(head_dim, eps=norm_eps, affine=True)
        self.k_norm = RmsNorm(head_dim, eps=norm_eps, affine=True)
        self.v_norm = RmsNorm(head_dim, eps=norm_eps, affine=False)  # no gain (HF omits it)

        self.attn_drop = nn.Dropout(attn_drop)
        self.proj_drop = nn.Dropout(proj_drop)

    def forward(self, x, rope_cos, rope_sin, attn_mask: Optional[torch.Tensor] = None):
        B, N, C = x.shape
        q = self.q_norm(self.q_proj(x).view(B, N, self.num_heads, self.head_dim))
        k = self.k_norm(self.k_proj(x).view(B, N, self.num_kv_heads, self.head_dim))
        v = self.v_norm(self.v_proj(x).view(B, N, self.num_kv_heads, self.head_dim))

        q = apply_multidimensional_rope(q, rope_cos, rope_sin, ndim=2, unsqueeze_dim=2)
        k = apply_multidimensional_rope(k, rope_cos, rope_sin, ndim=2, unsqueeze_dim=2)

        q = q.transpose(1, 2)
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        if self.num_kv_groups > 1:
            k = k.repeat_interleave(self.num_kv_groups, dim=1)
            v = v.repeat_interleave(self.num_kv_groups, dim=1)

        if self.fused_attn:
            x = ops.flash_attention(
                q, k, v, attn_mask=attn_mask,
                dropout_p=self.attn_drop.p if self.training else 0.0, scale=1.0)
        else:
            attn = q @ k.transpose(-2, -1)  # scale=1 (QK are normalized)
            if attn_mask is not None:
                attn = attn + attn_mask
            attn = self.attn_drop(attn.softmax(dim=-1))
            x = attn @ v

        x = x.transpose(1, 2).reshape(B, N, -1)
        x = self.o_proj(x)
        return self.proj_drop(x)


class Gemma4GatedMlp(nn.Module):
    """down(gelu_tanh(gate(x)) * up(x))."""

    def __init__(
            self,
            in_features: int,
            hidden_features: int,
            act_layer: Optional[Callable] = None,
            drop: float = 0.0,
            use_clipped_linears: bool = False,
    ):
        super().__init__()
        self.gate_proj = Gemma4ClippableLinear(in_features, hidden_features, use_clipped=use_clipped_linears)
        self.up_proj = Gemma4ClippableLinear(in_features, hidden_features, use_clipped=use_clipped_linears)
        self.down_proj = Gemma4ClippableLinear(hidden_features, in_features, use_clipped=use_clipped_linears)
        self.act = act_layer() if act_layer is not None else nn.GELU(approximate='tanh')
        self.drop = nn.Dropout(drop)

    def forward(self, x):
        return self.drop(self.down_proj(self.act(self.gate_proj(x)) * self.up_proj(x)))


class Gemma4Block(nn.Module):
    """4-norm sandwich block: norm-attn-norm + norm-mlp-norm residuals."""

    def __init__(
            self,
            dim: int,
            num_heads: int,
            head_dim: int,
            intermediate_size: int,
            num_kv_heads: Optional[int] = None,
            norm_eps: float = 1e-6,
            attn_drop: float = 0.0,
            proj_drop: float = 0.0,
            drop_path: float = 0.0,
            act_layer: Optional[Callable] = None,
            use_clipped_linears: bool = False,
    ):
        super().__init__()
        self.norm1 = RmsNorm(dim, eps=norm_eps)
        self.attn = Gemma4Attention(
            dim=dim, num_heads=num_heads, head_dim=head_dim, num_kv_heads=num_kv_heads,
            attn_drop=attn_drop, proj_drop=proj_drop, norm_eps=norm_eps,
            use_clipped_linears=use_clipped_linears)
        self.norm2 = RmsNorm(dim, eps=norm_eps)
        self.norm3 = RmsNorm(dim, eps=norm_eps)
        self.mlp = Gemma4GatedMlp(
            in_features=dim, hidden_features=intermediate_size,
            act_layer=act_layer, use_clipped_linears=use_clipped_linears)
        self.norm4 = RmsNorm(dim, eps=norm_eps)
        self.drop_path = DropPath(drop_path) if drop_path > 0.0 else nn.Identity()

    def forward(self, x, rope_cos, rope_sin, attn_mask: Optional[torch.Tensor] = None):
        residual = x
        x = self.norm2(self.attn(self.norm1(x), rope_cos, rope_sin, attn_mask=attn_mask))
        x = residual + self.drop_path(x)

        residual = x
        x = self.norm4(self.mlp(self.norm3(x)))
        x = residual + self.drop_path(x)
        return x


class Gemma4VisionPooler(nn.Module):
    """Average patches within k×k grid cells by position; ×√D scale."""

    def __init__(self, hidden_size: int, pooling_kernel_size: int = 3):
        super().__init__()
        self.hidden_size = hidden_size
        self.root_hidden_size = hidden_size ** 0.5
        self.pooling_kernel_size = pooling_kernel_size

    def _avg_pool_by_positions(self, hidden_states, position_ids):
        N = hidden_states.shape[1]
        k = self.pooling_kernel_size
        k2 = k * k
        if N % k2 != 0:
            raise ValueError(
                f'Cannot pool {N} tokens with k={k}: N must be divisible by k^2={k2}. '
                f'Both grid dimensions must be divisible by k.')
        out_len = N // k2

        clamped = position_ids.clamp(min=0)
        max_x = clamped[..., 0].max(dim=-1, keepdim=True)[0] + 1
        cell = torch.div(clamped, k, rounding_mode='floor')
        cell_idx = cell[..., 0] + (max_x // k) * cell[..., 1]

        weights = F.one_hot(cell_idx.long(), out_len).float() / k2
        out = weights.transpose(1, 2) @ hidden_states.float()
        mask = torch.logical_not((weights == 0).all(dim=1))
        return out.to(hidden_states.dtype), mask

    def forward(self, hidden_states, position_ids, padding_positions):
        hidden_states = hidden_states.masked_fill(padding_positions.unsqueeze(-1), 0.0)
        hidden_states, pooler_mask = self._avg_pool_by_positions(hidden_states, position_ids)
        return hidden_states * self.root_hidden_size, pooler_mask


class Gemma4VitEncoder(nn.Module):
    """Headless Gemma4 vision encoder (reference `gemma4_vit.py:618`)."""

    def __init__(
            self,
            img_size: Union[int, Tuple[int, int]] = 768,
            patch_size: int = 16,
            in_chans: int = 3,
            global_pool: str = 'soft',
            embed_dim: int = 768,
            depth: int = 16,
            num_heads: int = 12,
            head_dim: int = 64,
            num_kv_heads: Optional[int] = None,
            intermediate_size: int = 3072,
            norm_eps: float = 1e-6,
            rope_theta: float = 100.0,
            position_embedding_size: int = 10240,
            pooling_kernel_size: int = 3,
            standardize: bool = False,
            use_clipped_linears: bool = False,
            proj_drop_rate: float = 0.0,
            attn_drop_rate: float = 0.0,
            drop_path_rate: float = 0.0,
            act_layer: Optional[Callable] = None,
            weight_init: str = '',
    ):
        super().__init__()
        assert global_pool in ('soft', 'avg', 'none', '')
        self.global_pool = global_pool
        self.num_features = self.head_hidden_size = self.embed_dim = embed_dim
        self.num_classes = 0
        self.output_fmt = 'NLC'
        self.num_prefix_tokens = 0
        self.grad_checkpointing = False
        self.patch_size = to_2tuple(patch_size)
        self.pooling_kernel_size = pooling_kernel_size
        self.use_clipped_linears = use_clipped_linears

        act_layer = act_layer or partial(nn.GELU, approximate='tanh')

        self.patch_embed = Gemma4PatchEmbed(
            patch_size=self.patch_size, in_chans=in_chans, embed_dim=embed_dim,
            position_embedding_size=position_embedding_size)
        self.rotary_emb = Gemma4RotaryEmbedding2D(head_dim=head_dim, rope_theta=rope_theta)

        dpr = [x.item() for x in torch.linspace(0, drop_path_rate, depth)]
        self.blocks = nn.ModuleList([
            Gemma4Block(
                dim=embed_dim, num_heads=num_heads, head_dim=head_dim,
                num_kv_heads=num_kv_heads, intermediate_size=intermediate_size,
                norm_eps=norm_eps, attn_drop=attn_drop_rate, proj_drop=proj_drop_rate,
                drop_path=dpr[i], act_layer=act_layer, use_clipped_linears=use_clipped_linears)
            for i in range(depth)
        ])

        self.pooler = Gemma4VisionPooler(hidden_size=embed_dim, pooling_kernel_size=pooling_kernel_size)

        if standardize:
            # 31B variant's post-pool standardization; set by checkpoint
            self.register_buffer('std_bias', torch.empty(embed_dim))
            self.register_buffer('std_scale', torch.empty(embed_dim))
        else:
            self.std_bias = None
            self.std_scale = None

        _red = max(self.patch_size)
        self.feature_info = [dict(num_chs=embed_dim, reduction=_red, module=f'blocks.{i}') for i in range(depth)]

        self.weight_init_mode = 'reset' if weight_init == 'skip' else weight_init
        if weight_init != 'skip':
            self.init_weights(needs_reset=False)

    @torch.jit.ignore
    def init_weights(self, mode: str = '', needs_reset: bool = True) -> None:
        mode = mode or self.weight_init_mode
        assert mode in ('', 'reset')
        if self.std_bias is not None:
            nn.init.zeros_(self.std_bias)
        if self.std_scale is not None:
            nn.init.ones_(self.std_scale)
        named_apply(partial(init_weights_gemma4_vit, needs_reset=needs_reset), self)

    @torch.jit.ignore
    def no_weight_decay(self) -> Set[str]:
        return {'patch_embed.position_embedding_table'}

    @torch.jit.ignore
    def get_patch_size(self) -> Tuple[int, int]:
        return self.patch_size

    @torch.jit.ignore
    def group_matcher(self, coarse: bool = False) -> Dict[str, Any]:
        return dict(
            stem=r'^patch_embed|^rotary_emb',
            blocks=[(r'^blocks\.(\d+)', None), (r'^pooler|^std_', (99999,))],
        )

    @torch.jit.ignore
    def set_grad_checkpointing(self, enable: bool = True) -> None:
        self.grad_checkpointing = enable

    @torch.jit.ignore
    def set_clamp_enabled(self, enabled: bool = True) -> None:
        """Toggle the clamp ops on every Gemma4ClippableLinear (fine-tune aid)."""
        for mod in self.modules():
            if isinstance(mod, Gemma4ClippableLinear):
                mod.use_clipped = enabled

    def _assert_raw_img_conformant(self, x: torch.Tensor) -> None:
        if x.ndim != 4 or self.global_pool != 'soft':
            return
        H, W = x.shape[-2:]
        ph, pw = self.patch_size
        k = self.pooling_kernel_size
        if H % (ph * k) != 0 or W % (pw * k) != 0:
            raise ValueError(
                f"Image size ({H}, {W}) must be divisible by (patch_size * pooling_kernel_size) = "
                f"({ph * k}, {pw * k}) when global_pool='soft'.")

    def _encode(
            self,
            x: torch.Tensor,
            position_ids: torch.Tensor,
            padding_positions: torch.Tensor,
            block_callback: Optional[Callable[[int, torch.Tensor], None]] = None,
            max_block_index: Optional[int] = None,
    ) -> torch.Tensor:
        B, N = x.shape[:2]
        rope_cos, rope_sin = self.rotary_emb(x, position_ids)

        attn_mask: Optional[torch.Tensor] = None
        if padding_positions.any():
            attn_mask = torch.zeros(B, 1, 1, N, device=x.device, dtype=x.dtype)
            attn_mask.masked_fill_(padding_positions[:, None, None, :], float('-inf'))

        blocks = self.blocks if max_block_index is None else self.blocks[:max_block_index + 1]
        for i, blk in enumerate(blocks):
            if self.grad_checkpointing and not torch.jit.is_scripting():
                x = checkpoint(blk, x, rope_cos, rope_sin, attn_mask)
            else:
                x = blk(x, rope_cos, rope_sin, attn_mask=attn_mask)
            if block_callback is not None:
                block_callback(i, x)
        return x

    def forward_features(self, x, patch_coord=None, patch_valid=None) -> torch.Tensor:
        self._assert_raw_img_conformant(x if not isinstance(x, dict) else x['patches'])
        x, position_ids, padding_positions = self.patch_embed(x, patch_coord, patch_valid)
        return self._encode(x, position_ids, padding_positions)

    def forward_head(self, x: torch.Tensor, pre_logits: bool = False) -> torch.Tensor:
        raise NotImplementedError('Gemma4VitEncoder does not support classification use cases.')

    def forward(self, x, patch_coord=None, patch_valid=None) -> torch.Tensor:
        self._assert_raw_img_conformant(x if not isinstance(x, dict) else x['patches'])
        x, position_ids, padding_positions = self.patch_embed(x, patch_coord, patch_valid)
        x = self._encode(x, position_ids, padding_positions)

        if self.global_pool == 'soft':
            x, _ = self.pooler(x, position_ids, padding_positions)
            if self.std_bias is not None:
                x = (x - self.std_bias) * self.std_scale
        elif self.global_pool == 'avg':
            if padding_positions.any():
                x = x.masked_fill(padding_positions.unsqueeze(-1), 0.0)
                x = x.sum(dim=1) / (~padding_positions).sum(dim=1, keepdim=True).clamp(min=1)
            else:
                x = x.mean(dim=1)
        return x

    def forward_intermediates(
            self,
            x,
            patch_coord=None,
            patch_valid=None,
            indices: Optional[Union[int, List[int]]] = None,
            norm: bool = False,
            stop_early: bool = False,
            output_fmt: str = 'NCHW',
            intermediates_only: bool = False,
            output_dict: bool = False,
    ):
        assert output_fmt in ('NCHW', 'NLC')
        reshape = output_fmt == 'NCHW'
        take_indices, max_index = feature_take_indices(len(self.blocks), indices)

        raw = x if not isinstance(x, dict) else x['patches']
        raw_input_ndim = raw.ndim
        self._assert_raw_img_conformant(raw)
        x, position_ids, padding_positions = self.patch_embed(x, patch_coord, patch_valid)

        intermediates: List[torch.Tensor] = []

        def _cb(i: int, y: torch.Tensor) -> None:
            if i in take_indices:
                intermediates.append(y)

        max_block_index = max_index if (stop_early and not torch.jit.is_scripting()) else None
        x = self._encode(x, position_ids, padding_positions, block_callback=_cb, max_block_index=max_block_index)

        if reshape:
            if raw_input_ndim != 4:
                raise ValueError("output_fmt='NCHW' requires a raw image (B, C, H, W) input.")
            B = position_ids.shape[0]
            pW = int(position_ids[..., 0].max().item()) + 1
            pH = int(position_ids[..., 1].max().item()) + 1
            intermediates = [y.reshape(B, pH, pW, -1).permute(0, 3, 1, 2).contiguous() for y in intermediates]

        if output_dict:
            result: Dict[str, Any] = {'image_intermediates': intermediates}
            if not intermediates_only:
                result['image_features'] = x
            result['patch_valid'] = ~padding_positions
            return result
        if intermediates_only:
            return intermediates
        return x, intermediates

    def prune_intermediate_layers(self, indices=1, prune_norm: bool = False, prune_head: bool = True):
        take_indices, max_index = feature_take_indices(len(self.blocks), indices)
        self.blocks = self.blocks[:max_index + 1]
        return take_indices


class Gemma4VitClassifier(nn.Module):
    """Classification wrapper: encoder + optional RMSNorm + linear head."""

    def __init__(
            self,
            img_size: Union[int, Tuple[int, int]] = 768,
            patch_size: int = 16,
            in_chans: int = 3,
            num_classes: int = 1000,
            global_pool: str = 'avg',
            encoder_pool: str = '',
            embed_dim: int = 768,
            depth: int = 16,
            num_heads: int = 12,
            head_dim: int = 64,
            num_kv_heads: Optional[int] = None,
            intermediate_size: int = 3072,
            norm_eps: float = 1e-6,
            rope_theta: float = 100.0,
            position_embedding_size: int = 10240,
            pooling_kernel_size: int = 3,
            standardize: bool = False,
            use_clipped_linears: bool = False,
            final_norm: bool = True,
            drop_rate: float = 0.0,
            proj_drop_rate: float = 0.0,
            attn_drop_rate: float = 0.0,
            drop_path_rate: float = 0.0,
            act_layer: Optional[Callable] = None,
            weight_init: str = '',
    ):
        super().__init__()
        assert global_pool in ('avg', 'none', '')
        assert encoder_pool in ('', 'none', 'soft')
        self.num_classes = num_classes
        self.global_pool = global_pool
        self.encoder_pool = encoder_pool
        self.encoder = Gemma4VitEncoder(
            img_size=img_size, patch_size=patch_size, in_chans=in_chans,
            global_pool=encoder_pool, embed_dim=embed_dim, depth=depth,
            num_heads=num_heads, head_dim=head_dim, num_kv_heads=num_kv_heads,
            intermediate_size=intermediate_size, norm_eps=norm_eps,
            rope_theta=rope_theta, position_embedding_size=position_embedding_size,
            pooling_kernel_size=pooling_kernel_size, standardize=standardize,
            use_clipped_linears=use_clipped_linears, proj_drop_rate=proj_drop_rate,
            attn_drop_rate=attn_drop_rate, drop_path_rate=drop_path_rate,
            act_layer=act_layer, weight_init=weight_init)
        self.norm = RmsNorm(embed_dim, eps=norm_eps, affine=False) if final_norm else nn.Identity()
        self.head_drop = nn.Dropout(drop_rate)
        self.head = nn.Linear(embed_dim, num_classes) if num_classes > 0 else nn.Identity()

        self.num_features = self.head_hidden_size = self.encoder.num_features
        self.embed_dim = self.encoder.embed_dim
        self.patch_size = self.encoder.patch_size
        self.feature_info = self.encoder.feature_info

        self.weight_init_mode = self.encoder.weight_init_mode
        if weight_init != 'skip':
            if isinstance(self.head, nn.Linear) and self.head.bias is not None:
                nn.init.zeros_(self.head.bias)

    @torch.jit.ignore
    def init_weights(self, mode: str = '', needs_reset: bool = True) -> None:
        self.encoder.init_weights(mode=mode, needs_reset=needs_reset)
        if isinstance(self.head, nn.Linear) and self.head.bias is not None:
            nn.init.zeros_(self.head.bias)

    @torch.jit.ignore
    def no_weight_decay(self) -> Set[str]:
        return {f'encoder.{k}' for k in self.encoder.no_weight_decay()}

    @torch.jit.ignore
    def group_matcher(self, coarse: bool = False) -> Dict[str, Any]:
        return dict(
            stem=r'^encoder\.patch_embed|^encoder\.rotary_emb',
            blocks=[(r'^encoder\.blocks\.(\d+)', None)],
        )

    @torch.jit.ignore
    def set_grad_checkpointing(self, enable: bool = True) -> None:
        self.encoder.set_grad_checkpointing(enable)

    @torch.jit.ignore
    def set_clamp_enabled(self, enabled: bool = True) -> None:
        self.encoder.set_clamp_enabled(enabled)

    @torch.jit.ignore
    def get_patch_size(self) -> Tuple[int, int]:
        return self.encoder.get_patch_size()

    @torch.jit.ignore
    def get_classifier(self) -> nn.Module:
        return self.head

    def reset_classifier(self, num_classes: int, global_pool: Optional[str] = None) -> None:
        self.num_classes = num_classes
        if global_pool is not None:
            assert global_pool in ('avg', 'none', '')
            self.global_pool = global_pool
        self.head = nn.Linear(self.embed_dim, num_classes) if num_classes > 0 else nn.Identity()
        if isinstance(self.head, nn.Linear) and self.head.bias is not None:
            nn.init.zeros_(self.head.bias)

    def forward_features(self, x, patch_coord=None, patch_valid=None) -> torch.Tensor:
        if self.encoder_pool == 'soft':
            return self.encoder(x, patch_coord=patch_coord, patch_valid=patch_valid)
        return self.encoder.forward_features(x, patch_coord=patch_coord, patch_valid=patch_valid)

    def forward_head(self, x, patch_valid=None, pre_logits: bool = False) -> torch.Tensor:
        if self.global_pool == 'avg':
            if self.encoder_pool == 'soft' or patch_valid is None:
                x = x.mean(dim=1)
            else:
                x = x.masked_fill((~patch_valid).unsqueeze(-1), 0.0)
                x = x.sum(dim=1) / patch_valid.sum(dim=1, keepdim=True).clamp(min=1)
        x = self.norm(x)
        x = self.head_drop(x)
        return x if pre_logits else self.head(x)

    def forward(self, x, patch_coord=None, patch_valid=None) -> torch.Tensor:
        if isinstance(x, dict):
            patch_coord = x.get('patch_coord', patch_coord)
            patch_valid = x.get('patch_valid', patch_valid)
            x = x['patches']
        if patch_valid is None and patch_coord is not None:
            sentinel = (patch_coord == -1).all(dim=-1)
            if sentinel.any():
                patch_valid = ~sentinel

        feats = self.forward_features(x, patch_coord=patch_coord, patch_valid=patch_valid)
        return self.forward_head(feats, patch_valid=patch_valid)

    def forward_intermediates(
            self, x, patch_coord=None, patch_valid=None, indices=None,
            norm: bool = False, stop_early: bool = False,
            output_fmt: str = 'NCHW', intermediates_only: bool = False):
        return self.encoder.forward_intermediates(
            x, patch_coord=patch_coord, patch_valid=patch_valid, indices=indices,
            norm=norm, stop_early=stop_early, output_fmt=output_fmt,
            intermediates_only=intermediates_only)

    def prune_intermediate_layers(self, indices=1, prune_norm: bool = False, prune_head: bool = True):
        take_indices = self.encoder.prune_intermediate_layers(indices, prune_norm=prune_norm, prune_head=False)
        if prune_head:
            self.reset_classifier(0)
        return take_indices


def init_weights_gemma4_vit(module: nn.Module, name: str = '', needs_reset: bool = True) -> None:
    """Trunc-normal-TF for Linear weights; reset_parameters elsewhere."""
    if isinstance(module, nn.Linear):
        trunc_normal_tf_(module.weight, std=0.02)
        if module.bias is not None:
            nn.init.zeros_(module.bias)
    elif hasattr(module, 'init_weights'):
        module.init_weights()
    elif needs_reset and hasattr(module, 'reset_parameters'):
        module.reset_parameters()


def checkpoint_filter_fn_encoder(state_dict, model) -> Dict[str, torch.Tensor]:
    """HF Gemma4 vision encoder keys -> our encoder keys (pure renames)."""
    out_dict = {}
    hf_prefixes = ('model.vision_tower.', 'model.vision_model.', 'vision_model.', 'vision_tower.')

    for k, v in state_dict.items():
        matched = None
        for prefix in hf_prefixes:
            if k.startswith(prefix):
                matched = prefix
                break
        if matched is None:
            if k.startswith(('patch_embed.', 'blocks.', 'std_', 'pooler.', 'rotary_emb.')):
                out_dict[k] = v
            continue

        new_k = k[len(matched):]
        if 'rotary_emb' in new_k:
            continue  # recomputed buffers
        new_k = new_k.replace('patch_embedder.', 'patch_embed.')
        new_k = new_k.replace('encoder.layers.', 'blocks.')
        new_k = new_k.replace('.input_layernorm.', '.norm1.')
        new_k = new_k.replace('.post_attention_layernorm.', '.norm2.')
        new_k = new_k.replace('.pre_feedforward_layernorm.', '.norm3.')
        new_k = new_k.replace('.post_feedforward_layernorm.', '.norm4.')
        new_k = new_k.replace('.self_attn.', '.attn.')
        out_dict[new_k] = v
    return out_dict


def checkpoint_filter_fn_classifier(state_dict, model) -> Dict[str, torch.Tensor]:
    classifier_local = {k: v for k, v in state_dict.items()
                        if k.startswith(('norm.', 'head.', 'encoder.'))}
    to_filter = {k: v for k, v in state_dict.items() if k not in classifier_local}
    encoder_dict = checkpoint_filter_fn_encoder(to_filter, model.encoder)
    prefixed = {f'encoder.{k}': v for k, v in encoder_dict.items()}
    prefixed.update(classifier_local)
    return prefixed


def _create_gemma4_vit_encoder(variant: str, pretrained: bool = False, **kwargs) -> Gemma4VitEncoder:
    out_indices = kwargs.pop('out_indices', 3)
    return build_model_with_cfg(
        Gemma4VitEncoder, variant, pretrained,
        pretrained_filter_fn=checkpoint_filter_fn_encoder,
        feature_cfg=dict(out_indices=out_indices, feature_cls='getter'),
        kwargs_filter=('num_classes',),
        **kwargs,
    )


def _create_gemma4_vit_classifier(variant: str, pretrained: bool = False, **kwargs) -> Gemma4VitClassifier:
    out_indices = kwargs.pop('out_indices', 3)
    return build_model_with_cfg(
        Gemma4VitClassifier, variant, pretrained,
        pretrained_filter_fn=checkpoint_filter_fn_classifier,
        feature_cfg=dict(out_indices=out_indices, feature_cls='getter'),
        **kwargs,
    )


def _cfg(url: str = '', **kwargs) -> Dict[str, Any]:
    return {
        'url': url,
        'num_classes': 0,
        'input_size': (3, 768, 768),
        'min_input_size': (3, 96, 96),
        'pool_size': None,
        'crop_pct': 1.0,
        'interpolation': 'bicubic',
        'fixed_input_size': False,
        # the model scales [0,1] -> [-1,1] internally; pipeline passes raw [0,1]
        'mean': (0.0, 0.0, 0.0),
        'std': (1.0, 1.0, 1.0),
        'first_conv': 'patch_embed.input_proj',
        'classifier': 'head',
        **kwargs,
    }


default_cfgs = generate_default_cfgs({
    'gemma4_vit_167m.gemma4_e4b_it': _cfg(
        hf_hub_id='timm/gemma4_vit_167m.gemma4_e4b_it',
        first_conv='encoder.patch_embed.input_proj'),
    'gemma4_vit_570m.gemma4_31b_it': _cfg(
        hf_hub_id='timm/gemma4_vit_570m.gemma4_31b_it',
        first_conv='encoder.patch_embed.input_proj'),
    'gemma4_vit_167m_enc.gemma4_e4b_it': _cfg(hf_hub_id='timm/gemma4_vit_167m_enc.gemma4_e4b_it'),
    'gemma4_vit_570m_enc.gemma4_31b_it': _cfg(hf_hub_id='timm/gemma4_vit_570m_enc.gemma4_31b_it'),
})


_167M_ARCH = dict(
    embed_dim=768,
    depth=16,
    num_heads=12,
    head_dim=64,
    intermediate_size=3072,
    standardize=False,
    use_clipped_linears=True,
)

_570M_ARCH = dict(
    embed_dim=1152,
    depth=27,
    num_heads=16,
    head_dim=72,
    intermediate_size=4304,
    standardize=True,
)


@register_model
def gemma4_vit_167m(pretrained: bool = False, **kwargs) -> Gemma4VitClassifier:
    """Gemma4 ~167M (E2B/E4B vision tower) classifier."""
    model_args = dict(_167M_ARCH, final_norm=True)
    return _create_gemma4_vit_classifier('gemma4_vit_167m', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def gemma4_vit_167m_enc(pretrained: bool = False, **kwargs) -> Gemma4VitEncoder:
    """Gemma4 ~167M native VLM encoder (soft-token pool output)."""
    model_args = dict(_167M_ARCH, global_pool='soft')
    return _create_gemma4_vit_encoder('gemma4_vit_167m_enc', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def gemma4_vit_570m(pretrained: bool = False, **kwargs) -> Gemma4VitClassifier:
    """Gemma4 ~570M (26B/31B vision tower) classifier."""
    model_args = dict(_570M_ARCH, final_norm=True)
    return _create_gemma4_vit_classifier('gemma4_vit_570m', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def gemma4_vit_570m_enc(pretrained: bool = False, **kwargs) -> Gemma4VitEncoder:
    """Gemma4 ~570M native VLM encoder (soft-token pool + standardization)."""
    model_args = dict(_570M_ARCH, global_pool='soft')
    return _create_gemma4_vit_encoder('gemma4_vit_570m_enc', pretrained=pretrained, **dict(model_args, **kwargs))
