"""FastViT — hybrid rep-conv / attention vision transformer (Apple ML).

Capability parity with reference `timm/models/fastvit.py`: `MobileOneBlock`
(:43) with train-time overparameterized branches and `reparameterize()`
RepVGG-style BN folding, `ReparamLargeKernelConv` (:291, RepLKNet-style
large+small kernel fusion), RepMixer token mixer (:649), reparameterizable
conditional pos-enc (:823), conv-stem + 4/5-stage MetaFormer layout (:1171),
and the MobileCLIP (mci) variants.

MI355X notes: every reparameterizable block collapses to ONE dense conv at
deploy, ideal for MFMA implicit-GEMM; attention stages run at 1/32 scale
(small N) through our fused flash-attention path via layers.Attention.
"""
import os
from functools import partial
from typing import List, Optional, Tuple, Type, Union

import torch
import torch.nn as nn

from ..data.constants import IMAGENET_DEFAULT_MEAN, IMAGENET_DEFAULT_STD, OPENAI_CLIP_MEAN, OPENAI_CLIP_STD
from ..layers import (
    Attention, ClassifierHead, ConvNormAct, DropPath, LayerNorm2d, SqueezeExcite, create_conv2d, to_2tuple,
    trunc_normal_,
)
from ._builder import build_model_with_cfg
from ._features import feature_take_indices
from ._manipulate import checkpoint_seq
from ._registry import generate_default_cfgs, register_model

__all__ = ['FastVit', 'MobileOneBlock', 'ReparamLargeKernelConv']


def _num_groups(group_size: int, channels: int) -> int:
    if not group_size:
        return 1
    assert channels % group_size == 0
    return channels // group_size


def _fold_bn(conv_w: torch.Tensor, bn: nn.BatchNorm2d) -> Tuple[torch.Tensor, torch.Tensor]:
    """Fold BN stats into a conv weight: returns (w * gamma/std, beta - mean*gamma/std)."""
    std = (bn.running_var + bn.eps).sqrt()
    scale = (bn.weight / std).reshape(-1, 1, 1, 1)
    return conv_w * scale, bn.bias - bn.running_mean * bn.weight / std


def _identity_kernel(chs: int, groups: int, kernel_size: int, dtype, device) -> torch.Tensor:
    """Depthwise/grouped identity conv kernel (center tap = 1)."""
    in_dim = chs // groups
    w = torch.zeros(chs, in_dim, kernel_size, kernel_size, dtype=dtype, device=device)
    for i in range(chs):
        w[i, i % in_dim, kernel_size // 2, kernel_size // 2] = 1
    return w


class MobileOneBlock(nn.Module):
    """MobileOne block: multi-branch at train time, one conv at inference.

    Reference `fastvit.py:43`.  Branches: N kxk conv+BN, one 1x1 conv+BN
    scale branch, and a BN-only identity branch; `reparameterize()` folds
    all of them into `reparam_conv`.
    """

    def __init__(
            self,
            in_chs: int,
            out_chs: int,
            kernel_size: int,
            stride: int = 1,
            dilation: int = 1,
            group_size: int = 0,
            inference_mode: bool = False,
            use_se: bool = False,
            use_act: bool = True,
            use_scale_branch: bool = True,
            num_conv_branches: int = 1,
            act_layer: Type[nn.Module] = nn.GELU,
    ) -> None:
        super().__init__()
        self.inference_mode = inference_mode
        self.groups = _num_groups(group_size, in_chs)
        self.stride = stride
        self.dilation = dilation
        self.kernel_size = kernel_size
        self.in_chs = in_chs
        self.out_chs = out_chs
        self.num_conv_branches = num_conv_branches

        self.se = SqueezeExcite(out_chs, rd_divisor=1) if use_se else nn.Identity()

        if inference_mode:
            self.reparam_conv = create_conv2d(
                in_chs, out_chs, kernel_size=kernel_size, stride=stride,
                dilation=dilation, groups=self.groups, bias=True)
        else:
            self.reparam_conv = None
            self.identity = nn.BatchNorm2d(in_chs) if out_chs == in_chs and stride == 1 else None
            if num_conv_branches > 0:
                self.conv_kxk = nn.ModuleList([
                    ConvNormAct(
                        in_chs, out_chs, kernel_size=kernel_size, stride=stride,
                        groups=self.groups, apply_act=False)
                    for _ in range(num_conv_branches)
                ])
            else:
                self.conv_kxk = None
            self.conv_scale = None
            if kernel_size > 1 and use_scale_branch:
                self.conv_scale = ConvNormAct(
                    in_chs, out_chs, kernel_size=1, stride=stride, groups=self.groups, apply_act=False)

        self.act = act_layer() if use_act else nn.Identity()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.reparam_conv is not None:
            return self.act(self.se(self.reparam_conv(x)))

        out = 0
        if self.identity is not None:
            out = self.identity(x)
        if self.conv_scale is not None:
            out = out + self.conv_scale(x)
        if self.conv_kxk is not None:
            for branch in self.conv_kxk:
                out = out + branch(x)
        return self.act(self.se(out))

    def _branch_kernel_bias(self, branch) -> Tuple[torch.Tensor, torch.Tensor]:
        if isinstance(branch, ConvNormAct):
            return _fold_bn(branch.conv.weight, branch.bn)
        # BN-only identity branch
        assert isinstance(branch, nn.BatchNorm2d)
        if not hasattr(self, 'id_tensor'):
            self.id_tensor = _identity_kernel(
                self.in_chs, self.groups, self.kernel_size,
                branch.weight.dtype, branch.weight.device)
        return _fold_bn(self.id_tensor, branch)

    def _get_kernel_bias(self) -> Tuple[torch.Tensor, torch.Tensor]:
        w_sum, b_sum = 0, 0
        if self.conv_scale is not None:
            w, b = self._branch_kernel_bias(self.conv_scale)
            pad = self.kernel_size // 2
            w_sum = w_sum + nn.functional.pad(w, [pad] * 4)
            b_sum = b_sum + b
        if self.identity is not None:
            w, b = self._branch_kernel_bias(self.identity)
            w_sum, b_sum = w_sum + w, b_sum + b
        if self.conv_kxk is not None:
            for branch in self.conv_kxk:
                w, b = self._branch_kernel_bias(branch)
                w_sum, b_sum = w_sum + w, b_sum + b
        return w_sum, b_sum

    @torch.no_grad()
    def reparameterize(self):
        """Collapse all branches into a single conv for inference."""
        if self.reparam_conv is not None:
            return
        kernel, bias = self._get_kernel_bias()
        self.reparam_conv = create_conv2d(
            self.in_chs, self.out_chs, kernel_size=self.kernel_size, stride=self.stride,
            dilation=self.dilation, groups=self.groups, bias=True)
        self.reparam_conv.weight.data = kernel
        self.reparam_conv.bias.data = bias
        for name, p in self.named_parameters():
            if 'reparam_conv' not in name:
                p.detach_()
        self.__delattr__('conv_kxk')
        self.__delattr__('conv_scale')
        if hasattr(self, 'identity'):
            self.__delattr__('identity')
        self.inference_mode = True


class ReparamLargeKernelConv(nn.Module):
    """RepLKNet-style large-kernel conv with fusable parallel small kernel.

    Reference `fastvit.py:291`.
    """

    def __init__(
            self,
            in_chs: int,
            out_chs: int,
            kernel_size: int,
            stride: int,
            group_size: int,
            small_kernel: Optional[int] = None,
            use_se: bool = False,
            act_layer: Optional[Type[nn.Module]] = None,
            inference_mode: bool = False,
    ) -> None:
        super().__init__()
        self.stride = stride
        self.groups = _num_groups(group_size, in_chs)
        self.in_chs = in_chs
        self.out_chs = out_chs
        self.kernel_size = kernel_size
        self.small_kernel = small_kernel

        if inference_mode:
            self.reparam_conv = create_conv2d(
                in_chs, out_chs, kernel_size=kernel_size, stride=stride,
                dilation=1, groups=self.groups, bias=True)
        else:
            self.reparam_conv = None
            self.large_conv = ConvNormAct(
                in_chs, out_chs, kernel_size=kernel_size, stride=stride,
                groups=self.groups, apply_act=False)
            if small_kernel is not None:
                assert small_kernel <= kernel_size
                self.small_conv = ConvNormAct(
                    in_chs, out_chs, kernel_size=small_kernel, stride=stride,
                    groups=self.groups, apply_act=False)
        self.se = SqueezeExcite(out_chs, rd_ratio=0.25) if use_se else nn.Identity()
        self.act = act_layer() if act_layer is not None else nn.Identity()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.reparam_conv is not None:
            out = self.reparam_conv(x)
        else:
            out = self.large_conv(x)
            if self.small_conv is not None:
                out = out + self.small_conv(x)
        return self.act(self.se(out))

    def get_kernel_bias(self) -> Tuple[torch.Tensor, torch.Tensor]:
        w, b = _fold_bn(self.large_conv.conv.weight, self.large_conv.bn)
        if hasattr(self, 'small_conv'):
            ws, bs = _fold_bn(self.small_conv.conv.weight, self.small_conv.bn)
            b = b + bs
            w = w + nn.functional.pad(ws, [(self.kernel_size - self.small_kernel) // 2] * 4)
        return w, b

    @torch.no_grad()
    def reparameterize(self) -> None:
        kernel, bias = self.get_kernel_bias()
        self.reparam_conv = create_conv2d(
            self.in_chs, self.out_chs, kernel_size=self.kernel_size,
            stride=self.stride, groups=self.groups, bias=True)
        self.reparam_conv.weight.data = kernel
        self.reparam_conv.bias.data = bias
        self.__delattr__('large_conv')
        if hasattr(self, 'small_conv'):
            self.__delattr__('small_conv')


def convolutional_stem(
        in_chs: int,
        out_chs: int,
        act_layer: Type[nn.Module] = nn.GELU,
        inference_mode: bool = False,
        use_scale_branch: bool = True,
) -> nn.Sequential:
    """3-block MobileOne stem at stride 4 (reference `fastvit.py:449`)."""
    common = dict(act_layer=act_layer, inference_mode=inference_mode, use_scale_branch=use_scale_branch)
    return nn.Sequential(
        MobileOneBlock(in_chs, out_chs, kernel_size=3, stride=2, **common),
        MobileOneBlock(out_chs, out_chs, kernel_size=3, stride=2, group_size=1, **common),
        MobileOneBlock(out_chs, out_chs, kernel_size=1, stride=1, **common),
    )


class Attention2d(Attention):
    """MHSA token mixer over an NCHW map: flatten → shared Attention → restore.

    Reference `fastvit.py:504`; head_dim fixed (default 32).  Subclasses our
    shared Attention so parameter names (qkv/proj) match reference checkpoints.
    """

    def __init__(
            self,
            dim: int,
            head_dim: int = 32,
            qkv_bias: bool = False,
            attn_drop: float = 0.0,
            proj_drop: float = 0.0,
    ) -> None:
        assert dim % head_dim == 0
        super().__init__(
            dim, num_heads=dim // head_dim, qkv_bias=qkv_bias,
            attn_drop=attn_drop, proj_drop=proj_drop)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, C, H, W = x.shape
        x = x.flatten(2).transpose(-2, -1)
        x = super().forward(x)
        return x.transpose(-2, -1).reshape(B, C, H, W)


class PatchEmbed(nn.Module):
    """Stage downsample: reparam large-kernel conv + MobileOne 1x1 (reference `fastvit.py:575`)."""

    def __init__(
            self,
            patch_size: int,
            stride: int,
            in_chs: int,
            embed_dim: int,
            act_layer: Type[nn.Module] = nn.GELU,
            lkc_use_act: bool = False,
            use_se: bool = False,
            inference_mode: bool = False,
    ) -> None:
        super().__init__()
        self.proj = nn.Sequential(
            ReparamLargeKernelConv(
                in_chs, embed_dim, kernel_size=patch_size, stride=stride,
                group_size=1, small_kernel=3, use_se=use_se,
                act_layer=act_layer if lkc_use_act else None,
                inference_mode=inference_mode),
            MobileOneBlock(
                embed_dim, embed_dim, kernel_size=1, stride=1,
                act_layer=act_layer, inference_mode=inference_mode),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.proj(x)


class LayerScale2d(nn.Module):
    def __init__(self, dim: int, init_values: float = 1e-5, inplace: bool = False):
        super().__init__()
        self.inplace = inplace
        self.gamma = nn.Parameter(init_values * torch.ones(dim, 1, 1))

    def forward(self, x):
        return x.mul_(self.gamma) if self.inplace else x * self.gamma


class RepMixer(nn.Module):
    """Reparameterizable token mixer: x + ls * (mixer(x) - norm(x)) folds
    to one depthwise conv (reference `fastvit.py:649`)."""

    def __init__(
            self,
            dim: int,
            kernel_size: int = 3,
            layer_scale_init_value: Optional[float] = 1e-5,
            inference_mode: bool = False,
    ):
        super().__init__()
        self.dim = dim
        self.kernel_size = kernel_size
        self.inference_mode = inference_mode

        if inference_mode:
            self.reparam_conv = nn.Conv2d(
                dim, dim, kernel_size=kernel_size, stride=1,
                padding=kernel_size // 2, groups=dim, bias=True)
        else:
            self.reparam_conv = None
            self.norm = MobileOneBlock(
                dim, dim, kernel_size, group_size=1, use_act=False,
                use_scale_branch=False, num_conv_branches=0)
            self.mixer = MobileOneBlock(
                dim, dim, kernel_size, group_size=1, use_act=False)
            if layer_scale_init_value is not None:
                self.layer_scale = LayerScale2d(dim, layer_scale_init_value)
            else:
                self.layer_scale = nn.Identity()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.reparam_conv is not None:
            return self.reparam_conv(x)
        return x + self.layer_scale(self.mixer(x) - self.norm(x))

    @torch.no_grad()
    def reparameterize(self) -> None:
        if self.inference_mode:
            return
        self.mixer.reparameterize()
        self.norm.reparameterize()

        if isinstance(self.layer_scale, LayerScale2d):
            w = self.mixer.id_tensor + self.layer_scale.gamma.unsqueeze(-1) * (
                self.mixer.reparam_conv.weight - self.norm.reparam_conv.weight)
            b = torch.squeeze(self.layer_scale.gamma) * (
                self.mixer.reparam_conv.bias - self.norm.reparam_conv.bias)
        else:
            w = self.mixer.id_tensor + self.mixer.reparam_conv.weight - self.norm.reparam_conv.weight
            b = self.mixer.reparam_conv.bias - self.norm.reparam_conv.bias

        self.reparam_conv = create_conv2d(
            self.dim, self.dim, kernel_size=self.kernel_size, stride=1, groups=self.dim, bias=True)
        self.reparam_conv.weight.data = w
        self.reparam_conv.bias.data = b
        for name, p in self.named_parameters():
            if 'reparam_conv' not in name:
                p.detach_()
        self.__delattr__('mixer')
        self.__delattr__('norm')
        self.__delattr__('layer_scale')


class ConvMlp(nn.Module):
    """7x7 depthwise + 1x1 expand / project FFN (reference `fastvit.py:767`)."""

    def __init__(
            self,
            in_chs: int,
            hidden_channels: Optional[int] = None,
            out_chs: Optional[int] = None,
            act_layer: Type[nn.Module] = nn.GELU,
            drop: float = 0.0,
    ) -> None:
        super().__init__()
        out_chs = out_chs or in_chs
        hidden_channels = hidden_channels or in_chs
        self.conv = ConvNormAct(in_chs, out_chs, kernel_size=7, groups=in_chs, apply_act=False)
        self.fc1 = nn.Conv2d(in_chs, hidden_channels, kernel_size=1)
        self.act = act_layer()
        self.fc2 = nn.Conv2d(hidden_channels, out_chs, kernel_size=1)
        self.drop = nn.Dropout(drop)
        self.apply(self._init_weights)

    def _init_weights(self, m: nn.Module) -> None:
        if isinstance(m, nn.Conv2d):
            trunc_normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.conv(x)
        x = self.fc1(x)
        x = self.act(x)
        x = self.drop(x)
        x = self.fc2(x)
        x = self.drop(x)
        return x


class RepConditionalPosEnc(nn.Module):
    """Reparameterizable conditional positional encoding: depthwise conv +
    skip folds to one conv (reference `fastvit.py:823`)."""

    def __init__(
            self,
            dim: int,
            dim_out: Optional[int] = None,
            spatial_shape: Union[int, Tuple[int, int]] = (7, 7),
            inference_mode: bool = False,
    ) -> None:
        super().__init__()
        spatial_shape = to_2tuple(spatial_shape)
        self.spatial_shape = spatial_shape
        self.dim = dim
        self.dim_out = dim_out or dim
        self.groups = dim

        if inference_mode:
            self.reparam_conv = nn.Conv2d(
                dim, self.dim_out, kernel_size=spatial_shape, stride=1,
                padding=spatial_shape[0] // 2, groups=self.groups, bias=True)
        else:
            self.reparam_conv = None
            self.pos_enc = nn.Conv2d(
                dim, self.dim_out, spatial_shape, 1,
                spatial_shape[0] // 2, groups=self.groups, bias=True)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.reparam_conv is not None:
            return self.reparam_conv(x)
        return self.pos_enc(x) + x

    @torch.no_grad()
    def reparameterize(self) -> None:
        id_w = _identity_kernel(
            self.dim, self.groups, self.spatial_shape[0],
            self.pos_enc.weight.dtype, self.pos_enc.weight.device)
        self.reparam_conv = nn.Conv2d(
            self.dim, self.dim_out, kernel_size=self.spatial_shape, stride=1,
            padding=self.spatial_shape[0] // 2, groups=self.groups, bias=True)
        self.reparam_conv.weight.data = id_w + self.pos_enc.weight
        self.reparam_conv.bias.data = self.pos_enc.bias
        for name, p in self.named_parameters():
            if 'reparam_conv' not in name:
                p.detach_()
        self.__delattr__('pos_enc')


class RepMixerBlock(nn.Module):
    """MetaFormer block with RepMixer token mixer (reference `fastvit.py:944`)."""

    def __init__(
            self,
            dim: int,
            kernel_size: int = 3,
            mlp_ratio: float = 4.0,
            act_layer: Type[nn.Module] = nn.GELU,
            proj_drop: float = 0.0,
            drop_path: float = 0.0,
            layer_scale_init_value: Optional[float] = 1e-5,
            inference_mode: bool = False,
    ):
        super().__init__()
        self.token_mixer = RepMixer(
            dim, kernel_size=kernel_size,
            layer_scale_init_value=layer_scale_init_value,
            inference_mode=inference_mode)
        self.mlp = ConvMlp(dim, hidden_channels=int(dim * mlp_ratio), act_layer=act_layer, drop=proj_drop)
        if layer_scale_init_value is not None:
            self.layer_scale = LayerScale2d(dim, layer_scale_init_value)
        else:
            self.layer_scale = nn.Identity()
        self.drop_path = DropPath(drop_path) if drop_path > 0.0 else nn.Identity()

    def forward(self, x):
        x = self.token_mixer(x)
        x = x + self.drop_path(self.layer_scale(self.mlp(x)))
        return x


class AttentionBlock(nn.Module):
    """MetaFormer block with MHSA token mixer (reference `fastvit.py:1006`)."""

    def __init__(
            self,
            dim: int,
            mlp_ratio: float = 4.0,
            act_layer: Type[nn.Module] = nn.GELU,
            norm_layer: Type[nn.Module] = nn.BatchNorm2d,
            proj_drop: float = 0.0,
            drop_path: float = 0.0,
            layer_scale_init_value: Optional[float] = 1e-5,
    ):
        super().__init__()
        self.norm = norm_layer(dim)
        self.token_mixer = Attention2d(dim=dim)
        if layer_scale_init_value is not None:
            self.layer_scale_1 = LayerScale2d(dim, layer_scale_init_value)
        else:
            self.layer_scale_1 = nn.Identity()
        self.drop_path1 = DropPath(drop_path) if drop_path > 0.0 else nn.Identity()

        self.mlp = ConvMlp(dim, hidden_channels=int(dim * mlp_ratio), act_layer=act_layer, drop=proj_drop)
        if layer_scale_init_value is not None:
            self.layer_scale_2 = LayerScale2d(dim, layer_scale_init_value)
        else:
            self.layer_scale_2 = nn.Identity()
        self.drop_path2 = DropPath(drop_path) if drop_path > 0.0 else nn.Identity()

    def forward(self, x):
        x = x + self.drop_path1(self.layer_scale_1(self.token_mixer(self.norm(x))))
        x = x + self.drop_path2(self.layer_scale_2(self.mlp(x)))
        return x


class FastVitStage(nn.Module):
    """Downsample + optional pos-enc + mixer blocks (reference `fastvit.py:1066`)."""

    def __init__(
            self,
            dim: int,
            dim_out: int,
            depth: int,
            token_mixer_type: str,
            downsample: bool = True,
            se_downsample: bool = False,
            down_patch_size: int = 7,
            down_stride: int = 2,
            pos_emb_layer: Optional[Type[nn.Module]] = None,
            kernel_size: int = 3,
            mlp_ratio: float = 4.0,
            act_layer: Type[nn.Module] = nn.GELU,
            norm_layer: Type[nn.Module] = nn.BatchNorm2d,
            proj_drop_rate: float = 0.0,
            drop_path_rate: Union[List[float], float] = 0.0,
            layer_scale_init_value: Optional[float] = 1e-5,
            lkc_use_act: bool = False,
            inference_mode: bool = False,
    ):
        super().__init__()
        self.grad_checkpointing = False

        if downsample:
            self.downsample = PatchEmbed(
                patch_size=down_patch_size, stride=down_stride, in_chs=dim,
                embed_dim=dim_out, use_se=se_downsample, act_layer=act_layer,
                lkc_use_act=lkc_use_act, inference_mode=inference_mode)
        else:
            assert dim == dim_out
            self.downsample = nn.Identity()

        if pos_emb_layer is not None:
            self.pos_emb = pos_emb_layer(dim_out, inference_mode=inference_mode)
        else:
            self.pos_emb = nn.Identity()

        if isinstance(drop_path_rate, (int, float)):
            drop_path_rate = [drop_path_rate] * depth
        blocks = []
        for block_idx in range(depth):
            if token_mixer_type == 'repmixer':
                blocks.append(RepMixerBlock(
                    dim_out, kernel_size=kernel_size, mlp_ratio=mlp_ratio,
                    act_layer=act_layer, proj_drop=proj_drop_rate,
                    drop_path=drop_path_rate[block_idx],
                    layer_scale_init_value=layer_scale_init_value,
                    inference_mode=inference_mode))
            elif token_mixer_type == 'attention':
                blocks.append(AttentionBlock(
                    dim_out, mlp_ratio=mlp_ratio, act_layer=act_layer,
                    norm_layer=norm_layer, proj_drop=proj_drop_rate,
                    drop_path=drop_path_rate[block_idx],
                    layer_scale_init_value=layer_scale_init_value))
            else:
                raise ValueError(f'Token mixer type: {token_mixer_type} not supported')
        self.blocks = nn.Sequential(*blocks)

    def forward(self, x):
        x = self.downsample(x)
        x = self.pos_emb(x)
        if self.grad_checkpointing and not torch.jit.is_scripting():
            x = checkpoint_seq(self.blocks, x)
        else:
            x = self.blocks(x)
        return x


class FastVit(nn.Module):
    """FastViT (reference `fastvit.py:1171`)."""
    fork_feat: torch.jit.Final[bool]

    def __init__(
            self,
            in_chans: int = 3,
            layers: Tuple[int, ...] = (2, 2, 6, 2),
            token_mixers: Tuple[str, ...] = ('repmixer',) * 4,
            embed_dims: Tuple[int, ...] = (64, 128, 256, 512),
            mlp_ratios: Tuple[float, ...] = (4,) * 4,
            downsamples: Tuple[bool, ...] = (False, True, True, True),
            se_downsamples: Tuple[bool, ...] = (False, False, False, False),
            repmixer_kernel_size: int = 3,
            num_classes: int = 1000,
            pos_embs: Tuple[Optional[Type[nn.Module]], ...] = (None,) * 4,
            down_patch_size: int = 7,
            down_stride: int = 2,
            drop_rate: float = 0.0,
            proj_drop_rate: float = 0.0,
            drop_path_rate: float = 0.0,
            layer_scale_init_value: float = 1e-5,
            lkc_use_act: bool = False,
            stem_use_scale_branch: bool = True,
            fork_feat: bool = False,
            cls_ratio: float = 2.0,
            global_pool: str = 'avg',
            norm_layer: Type[nn.Module] = nn.BatchNorm2d,
            act_layer: Type[nn.Module] = nn.GELU,
            inference_mode: bool = False,
    ) -> None:
        super().__init__()
        self.num_classes = 0 if fork_feat else num_classes
        self.fork_feat = fork_feat
        self.global_pool = global_pool
        self.feature_info = []

        self.stem = convolutional_stem(
            in_chans, embed_dims[0], act_layer, inference_mode,
            use_scale_branch=stem_use_scale_branch)

        prev_dim = embed_dims[0]
        scale = 1
        dpr = [x.tolist() for x in torch.linspace(0, drop_path_rate, sum(layers)).split(list(layers))]
        stages = []
        for i in range(len(layers)):
            downsample = downsamples[i] or prev_dim != embed_dims[i]
            stage = FastVitStage(
                dim=prev_dim,
                dim_out=embed_dims[i],
                depth=layers[i],
                downsample=downsample,
                se_downsample=se_downsamples[i],
                down_patch_size=down_patch_size,
                down_stride=down_stride,
                pos_emb_layer=pos_embs[i],
                token_mixer_type=token_mixers[i],
                kernel_size=repmixer_kernel_size,
                mlp_ratio=mlp_ratios[i],
                act_layer=act_layer,
                norm_layer=norm_layer,
                proj_drop_rate=proj_drop_rate,
                drop_path_rate=dpr[i],
                layer_scale_init_value=layer_scale_init_value,
                lkc_use_act=lkc_use_act,
                inference_mode=inference_mode,
            )
            stages.append(stage)
            prev_dim = embed_dims[i]
            if downsample:
                scale *= 2
            self.feature_info += [dict(num_chs=prev_dim, reduction=4 * scale, module=f'stages.{i}')]
        self.stages = nn.Sequential(*stages)
        self.num_stages = len(self.stages)
        self.num_features = self.head_hidden_size = prev_dim

        if self.fork_feat:
            self.out_indices = [0, 1, 2, 3]
            for i_emb, i_layer in enumerate(self.out_indices):
                if i_emb == 0 and os.environ.get('FORK_LAST3', None):
                    layer = nn.Identity()
                else:
                    layer = norm_layer(embed_dims[i_emb])
                self.add_module(f'norm{i_layer}', layer)
            self.final_conv = nn.Identity()
            self.head = nn.Identity()
        else:
            self.num_features = self.head_hidden_size = final_features = int(embed_dims[-1] * cls_ratio)
            self.final_conv = MobileOneBlock(
                embed_dims[-1], final_features, kernel_size=3, stride=1, group_size=1,
                inference_mode=inference_mode, use_se=True, act_layer=act_layer,
                num_conv_branches=1)
            self.head = ClassifierHead(
                final_features, num_classes, pool_type=global_pool, drop_rate=drop_rate)

        self.apply(self._init_weights)

    def _init_weights(self, m: nn.Module) -> None:
        if isinstance(m, nn.Linear):
            trunc_normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)

    @torch.jit.ignore
    def no_weight_decay(self):
        return set()

    @torch.jit.ignore
    def group_matcher(self, coarse=False):
        return dict(
            stem=r'^stem',
            blocks=r'^stages\.(\d+)' if coarse else [
                (r'^stages\.(\d+).downsample', (0,)),
                (r'^stages\.(\d+).pos_emb', (0,)),
                (r'^stages\.(\d+)\.\w+\.(\d+)', None),
            ]
        )

    @torch.jit.ignore
    def set_grad_checkpointing(self, enable=True):
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
        last_idx = self.num_stages - 1
        if torch.jit.is_scripting() or not stop_early:
            stages = self.stages
        else:
            stages = self.stages[:max_index + 1]
        feat_idx = 0
        for feat_idx, stage in enumerate(stages):
            x = stage(x)
            if feat_idx in take_indices:
                intermediates.append(x)

        if intermediates_only:
            return intermediates

        if feat_idx == last_idx:
            x = self.final_conv(x)

        return x, intermediates

    def prune_intermediate_layers(
            self,
            indices: Union[int, List[int]] = 1,
            prune_norm: bool = False,
            prune_head: bool = True,
    ):
        take_indices, max_index = feature_take_indices(len(self.stages), indices)
        self.stages = self.stages[:max_index + 1]
        if prune_head:
            self.reset_classifier(0, '')
        return take_indices

    def forward_features(self, x: torch.Tensor) -> torch.Tensor:
        x = self.stem(x)
        outs = []
        for idx, stage in enumerate(self.stages):
            x = stage(x)
            if self.fork_feat and idx in self.out_indices:
                norm_layer = getattr(self, f'norm{idx}')
                outs.append(norm_layer(x))
        if self.fork_feat:
            return outs
        x = self.final_conv(x)
        return x

    def forward_head(self, x: torch.Tensor, pre_logits: bool = False):
        return self.head(x, pre_logits=True) if pre_logits else self.head(x)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.forward_features(x)
        if self.fork_feat:
            return x
        x = self.forward_head(x)
        return x


def _cfg(url='', **kwargs):
    return {
        'url': url,
        'num_classes': 1000,
        'input_size': (3, 256, 256),
        'pool_size': (8, 8),
        'crop_pct': 0.9,
        'interpolation': 'bicubic',
        'mean': IMAGENET_DEFAULT_MEAN,
        'std': IMAGENET_DEFAULT_STD,
        'first_conv': ('stem.0.conv_kxk.0.conv', 'stem.0.conv_scale.conv'),
        'classifier': 'head.fc',
        **kwargs,
    }


default_cfgs = generate_default_cfgs({
    'fastvit_t8.apple_in1k': _cfg(hf_hub_id='timm/'),
    'fastvit_t12.apple_in1k': _cfg(hf_hub_id='timm/'),
    'fastvit_s12.apple_in1k': _cfg(hf_hub_id='timm/'),
    'fastvit_sa12.apple_in1k': _cfg(hf_hub_id='timm/'),
    'fastvit_sa24.apple_in1k': _cfg(hf_hub_id='timm/'),
    'fastvit_sa36.apple_in1k': _cfg(hf_hub_id='timm/'),
    'fastvit_ma36.apple_in1k': _cfg(hf_hub_id='timm/', crop_pct=0.95),

    'fastvit_t8.apple_dist_in1k': _cfg(hf_hub_id='timm/'),
    'fastvit_t12.apple_dist_in1k': _cfg(hf_hub_id='timm/'),
    'fastvit_s12.apple_dist_in1k': _cfg(hf_hub_id='timm/'),
    'fastvit_sa12.apple_dist_in1k': _cfg(hf_hub_id='timm/'),
    'fastvit_sa24.apple_dist_in1k': _cfg(hf_hub_id='timm/'),
    'fastvit_sa36.apple_dist_in1k': _cfg(hf_hub_id='timm/'),
    'fastvit_ma36.apple_dist_in1k': _cfg(hf_hub_id='timm/', crop_pct=0.95),

    'fastvit_mci0.apple_mclip': _cfg(
        hf_hub_id='apple/mobileclip_s0_timm', crop_pct=0.95,
        num_classes=512, mean=(0., 0., 0.), std=(1., 1., 1.)),
    'fastvit_mci1.apple_mclip': _cfg(
        hf_hub_id='apple/mobileclip_s1_timm', crop_pct=0.95,
        num_classes=512, mean=(0., 0., 0.), std=(1., 1., 1.)),
    'fastvit_mci2.apple_mclip': _cfg(
        hf_hub_id='apple/mobileclip_s2_timm', crop_pct=0.95,
        num_classes=512, mean=(0., 0., 0.), std=(1., 1., 1.)),
    'fastvit_mci0.apple_mclip2_dfndr2b': _cfg(
        hf_hub_id='timm/', crop_pct=1.0,
        num_classes=512, mean=(0., 0., 0.), std=(1., 1., 1.)),
    'fastvit_mci2.apple_mclip2_dfndr2b': _cfg(
        hf_hub_id='timm/', crop_pct=0.95,
        num_classes=512, mean=(0., 0., 0.), std=(1., 1., 1.)),
    'fastvit_mci3.apple_mclip2_dfndr2b': _cfg(
        hf_hub_id='timm/', crop_pct=0.95, num_classes=768,
        mean=OPENAI_CLIP_MEAN, std=OPENAI_CLIP_STD, pool_size=(4, 4),
        first_conv='stem.0.conv_kxk.0.conv'),
    'fastvit_mci4.apple_mclip2_dfndr2b': _cfg(
        hf_hub_id='timm/', crop_pct=0.95, num_classes=768,
        mean=OPENAI_CLIP_MEAN, std=OPENAI_CLIP_STD, pool_size=(4, 4),
        first_conv='stem.0.conv_kxk.0.conv'),
})


def _create_fastvit(variant, pretrained=False, **kwargs):
    out_indices = kwargs.pop('out_indices', (0, 1, 2, 3))
    return build_model_with_cfg(
        FastVit,
        variant,
        pretrained,
        feature_cfg=dict(flatten_sequential=True, out_indices=out_indices),
        **kwargs,
    )


@register_model
def fastvit_t8(pretrained=False, **kwargs):
    """FastViT-T8."""
    model_args = dict(
        layers=(2, 2, 4, 2),
        embed_dims=(48, 96, 192, 384),
        mlp_ratios=(3, 3, 3, 3),
        token_mixers=('repmixer',) * 4,
    )
    return _create_fastvit('fastvit_t8', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def fastvit_t12(pretrained=False, **kwargs):
    """FastViT-T12."""
    model_args = dict(
        layers=(2, 2, 6, 2),
        embed_dims=(64, 128, 256, 512),
        mlp_ratios=(3, 3, 3, 3),
        token_mixers=('repmixer',) * 4,
    )
    return _create_fastvit('fastvit_t12', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def fastvit_s12(pretrained=False, **kwargs):
    """FastViT-S12."""
    model_args = dict(
        layers=(2, 2, 6, 2),
        embed_dims=(64, 128, 256, 512),
        mlp_ratios=(4, 4, 4, 4),
        token_mixers=('repmixer',) * 4,
    )
    return _create_fastvit('fastvit_s12', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def fastvit_sa12(pretrained=False, **kwargs):
    """FastViT-SA12 (attention final stage)."""
    model_args = dict(
        layers=(2, 2, 6, 2),
        embed_dims=(64, 128, 256, 512),
        mlp_ratios=(4, 4, 4, 4),
        pos_embs=(None, None, None, partial(RepConditionalPosEnc, spatial_shape=(7, 7))),
        token_mixers=('repmixer', 'repmixer', 'repmixer', 'attention'),
    )
    return _create_fastvit('fastvit_sa12', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def fastvit_sa24(pretrained=False, **kwargs):
    """FastViT-SA24."""
    model_args = dict(
        layers=(4, 4, 12, 4),
        embed_dims=(64, 128, 256, 512),
        mlp_ratios=(4, 4, 4, 4),
        pos_embs=(None, None, None, partial(RepConditionalPosEnc, spatial_shape=(7, 7))),
        token_mixers=('repmixer', 'repmixer', 'repmixer', 'attention'),
    )
    return _create_fastvit('fastvit_sa24', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def fastvit_sa36(pretrained=False, **kwargs):
    """FastViT-SA36."""
    model_args = dict(
        layers=(6, 6, 18, 6),
        embed_dims=(64, 128, 256, 512),
        mlp_ratios=(4, 4, 4, 4),
        pos_embs=(None, None, None, partial(RepConditionalPosEnc, spatial_shape=(7, 7))),
        token_mixers=('repmixer', 'repmixer', 'repmixer', 'attention'),
    )
    return _create_fastvit('fastvit_sa36', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def fastvit_ma36(pretrained=False, **kwargs):
    """FastViT-MA36."""
    model_args = dict(
        layers=(6, 6, 18, 6),
        embed_dims=(76, 152, 304, 608),
        mlp_ratios=(4, 4, 4, 4),
        pos_embs=(None, None, None, partial(RepConditionalPosEnc, spatial_shape=(7, 7))),
        token_mixers=('repmixer', 'repmixer', 'repmixer', 'attention'),
    )
    return _create_fastvit('fastvit_ma36', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def fastvit_mci0(pretrained=False, **kwargs):
    """MobileCLIP image tower MCi0."""
    model_args = dict(
        layers=(2, 6, 10, 2),
        embed_dims=(64, 128, 256, 512),
        mlp_ratios=(3, 3, 3, 3),
        se_downsamples=(False, False, True, True),
        pos_embs=(None, None, None, partial(RepConditionalPosEnc, spatial_shape=(7, 7))),
        token_mixers=('repmixer', 'repmixer', 'repmixer', 'attention'),
        lkc_use_act=True,
    )
    return _create_fastvit('fastvit_mci0', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def fastvit_mci1(pretrained=False, **kwargs):
    """MobileCLIP image tower MCi1."""
    model_args = dict(
        layers=(4, 12, 20, 4),
        embed_dims=(64, 128, 256, 512),
        mlp_ratios=(3, 3, 3, 3),
        se_downsamples=(False, False, True, True),
        pos_embs=(None, None, None, partial(RepConditionalPosEnc, spatial_shape=(7, 7))),
        token_mixers=('repmixer', 'repmixer', 'repmixer', 'attention'),
        lkc_use_act=True,
    )
    return _create_fastvit('fastvit_mci1', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def fastvit_mci2(pretrained=False, **kwargs):
    """MobileCLIP image tower MCi2."""
    model_args = dict(
        layers=(4, 12, 24, 4),
        embed_dims=(80, 160, 320, 640),
        mlp_ratios=(3, 3, 3, 3),
        se_downsamples=(False, False, True, True),
        pos_embs=(None, None, None, partial(RepConditionalPosEnc, spatial_shape=(7, 7))),
        token_mixers=('repmixer', 'repmixer', 'repmixer', 'attention'),
        lkc_use_act=True,
    )
    return _create_fastvit('fastvit_mci2', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def fastvit_mci3(pretrained=False, **kwargs):
    """MobileCLIP2 image tower L (5 stages)."""
    model_args = dict(
        layers=(2, 12, 24, 4, 2),
        embed_dims=(96, 192, 384, 768, 1536),
        mlp_ratios=(4, 4, 4, 4, 4),
        se_downsamples=(False,) * 5,
        downsamples=(False, True, True, True, True),
        pos_embs=(
            None, None, None,
            partial(RepConditionalPosEnc, spatial_shape=(7, 7)),
            partial(RepConditionalPosEnc, spatial_shape=(7, 7)),
        ),
        token_mixers=('repmixer', 'repmixer', 'repmixer', 'attention', 'attention'),
        lkc_use_act=True,
        norm_layer=partial(LayerNorm2d, eps=1e-5),
        stem_use_scale_branch=False,
    )
    return _create_fastvit('fastvit_mci3', pretrained=pretrained, **dict(model_args, **kwargs))


@register_model
def fastvit_mci4(pretrained=False, **kwargs):
    """MobileCLIP2 image tower XL (5 stages)."""
    model_args = dict(
        layers=(2, 12, 24, 4, 4),
        embed_dims=(128, 256, 512, 1024, 2048),
        mlp_ratios=(4, 4, 4, 4, 4),
        se_downsamples=(False,) * 5,
        downsamples=(False, True, True, True, True),
        pos_embs=(
            None, None, None,
            partial(RepConditionalPosEnc, spatial_shape=(7, 7)),
            partial(RepConditionalPosEnc, spatial_shape=(7, 7)),
        ),
        token_mixers=('repmixer', 'repmixer', 'repmixer', 'attention', 'attention'),
        lkc_use_act=True,
        norm_layer=partial(LayerNorm2d, eps=1e-5),
        stem_use_scale_branch=False,
    )
    return _create_fastvit('fastvit_mci4', pretrained=pretrained, **dict(model_args, **kwargs))
