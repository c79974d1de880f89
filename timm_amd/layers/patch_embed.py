"""Patch embedding (reference `timm/layers/patch_embed.py:26-142`).

The stride-p conv is mathematically a per-patch GEMM; on device we lower it to
an im2row-free reshape + hipBLASLt GEMM (`_patchify_gemm`) instead of going
through MIOpen conv — patches are contiguous blocks so the reshape is pure
view manipulation (no gather kernel).
"""
import logging
import math
from typing import Callable, List, Optional, Tuple, Union

import torch
import torch.nn.functional as F
from torch import nn

from .format import Format, nchw_to
from .helpers import to_2tuple
from .trace_utils import _assert

_logger = logging.getLogger(__name__)


def _patchify_gemm(x: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor],
                   patch_size: Tuple[int, int]) -> torch.Tensor:
    """[B,C,H,W] -> [B, nH*nW, D] patch-embed via reshape + GEMM (stride == kernel)."""
    B, C, H, W = x.shape
    ph, pw = patch_size
    nh, nw = H // ph, W // pw
    # [B, C, nh, ph, nw, pw] -> [B, nh, nw, C, ph, pw] -> [B, nh*nw, C*ph*pw]
    x = x.view(B, C, nh, ph, nw, pw).permute(0, 2, 4, 1, 3, 5).reshape(B, nh * nw, C * ph * pw)
    w = weight.reshape(weight.shape[0], -1)  # [D, C*ph*pw]
    return F.linear(x, w, bias)


class PatchEmbed(nn.Module):
    """2D Image to Patch Embedding."""
    output_fmt: Format
    dynamic_img_pad: torch.jit.Final[bool]

    def __init__(
            self,
            img_size: Optional[int] = 224,
            patch_size: int = 16,
            in_chans: int = 3,
            embed_dim: int = 768,
            norm_layer: Optional[Callable] = None,
            flatten: bool = True,
            output_fmt: Optional[str] = None,
            bias: bool = True,
            strict_img_size: bool = True,
            dynamic_img_pad: bool = False,
    ):
        super().__init__()
        self.patch_size = to_2tuple(patch_size)
        self.img_size, self.grid_size, self.num_patches = self._init_img_size(img_size)

        if output_fmt is not None:
            self.flatten = False
            self.output_fmt = Format(output_fmt)
        else:
            # flatten spatial dim and transpose to channels last, kept for bwd compat
            self.flatten = flatten
            self.output_fmt = Format.NCHW
        self.strict_img_size = strict_img_size
        self.dynamic_img_pad = dynamic_img_pad

        self.proj = nn.Conv2d(in_chans, embed_dim, kernel_size=patch_size, stride=patch_size, bias=bias)
        self.norm = norm_layer(embed_dim) if norm_layer else nn.Identity()

    def _init_img_size(self, img_size: Union[int, Tuple[int, int], None]):
        if img_size is None:
            return None, None, None
        img_size = to_2tuple(img_size)
        grid_size = tuple(s // p for s, p in zip(img_size, self.patch_size))
        num_patches = grid_size[0] * grid_size[1]
        return img_size, grid_size, num_patches

    def set_input_size(self, img_size=None, patch_size=None):
        new_patch_size = None
        if patch_size is not None:
            new_patch_size = to_2tuple(patch_size)
        if new_patch_size is not None and new_patch_size != self.patch_size:
            with torch.no_grad():
                new_proj = nn.Conv2d(
                    self.proj.in_channels, self.proj.out_channels,
                    kernel_size=new_patch_size, stride=new_patch_size,
                    bias=self.proj.bias is not None,
                )
                new_proj.weight.copy_(resample_patch_embed(self.proj.weight, list(new_patch_size), verbose=True))
                if self.proj.bias is not None:
                    new_proj.bias.copy_(self.proj.bias)
                self.proj = new_proj
            self.patch_size = new_patch_size
        img_size = img_size or self.img_size
        if img_size != self.img_size or new_patch_size is not None:
            self.img_size, self.grid_size, self.num_patches = self._init_img_size(img_size)

    def feat_ratio(self, as_scalar=True) -> Union[Tuple[int, int], int]:
        if as_scalar:
            return max(self.patch_size)
        return self.patch_size

    def dyn_feat_size(self, img_size: Tuple[int, int]) -> Tuple[int, int]:
        """Expected feature size for given image size, taking dynamic padding into account."""
        if self.dynamic_img_pad:
            return math.ceil(img_size[0] / self.patch_size[0]), math.ceil(img_size[1] / self.patch_size[1])
        return img_size[0] // self.patch_size[0], img_size[1] // self.patch_size[1]

    def forward(self, x):
        B, C, H, W = x.shape
        if self.img_size is not None:
            if self.strict_img_size:
                _assert(H == self.img_size[0], f"Input height ({H}) doesn't match model ({self.img_size[0]}).")
                _assert(W == self.img_size[1], f"Input width ({W}) doesn't match model ({self.img_size[1]}).")
            elif not self.dynamic_img_pad:
                _assert(H % self.patch_size[0] == 0, f"Input height ({H}) should be divisible by patch size.")
                _assert(W % self.patch_size[1] == 0, f"Input width ({W}) should be divisible by patch size.")
        if self.dynamic_img_pad:
            pad_h = (self.patch_size[0] - H % self.patch_size[0]) % self.patch_size[0]
            pad_w = (self.patch_size[1] - W % self.patch_size[1]) % self.patch_size[1]
            x = F.pad(x, (0, pad_w, 0, pad_h))
            H, W = H + pad_h, W + pad_w

        if self.flatten:
            # GEMM path yields NLC directly
            x = _patchify_gemm(x, self.proj.weight, self.proj.bias, self.patch_size)
        else:
            nh, nw = H // self.patch_size[0], W // self.patch_size[1]
            x = _patchify_gemm(x, self.proj.weight, self.proj.bias, self.patch_size)
            x = x.view(B, nh, nw, -1)
            if self.output_fmt == Format.NCHW:
                x = x.permute(0, 3, 1, 2)
            elif self.output_fmt == Format.NLC:
                x = x.flatten(1, 2)
        x = self.norm(x)
        return x


class PatchEmbedWithSize(PatchEmbed):
    """2D Image to Patch Embedding that also returns the feature grid size."""

    def forward(self, x) -> Tuple[torch.Tensor, List[int]]:
        B, C, H, W = x.shape
        if self.img_size is not None:
            _assert(H % self.patch_size[0] == 0, f"Input image height ({H}) must be divisible by patch size ({self.patch_size[0]}).")
            _assert(W % self.patch_size[1] == 0, f"Input image width ({W}) must be divisible by patch size ({self.patch_size[1]}).")
        feat_size = (H // self.patch_size[0], W // self.patch_size[1])
        x = _patchify_gemm(x, self.proj.weight, self.proj.bias, self.patch_size)
        if not self.flatten:
            x = x.view(B, feat_size[0], feat_size[1], -1)
            if self.output_fmt == Format.NCHW:
                x = x.permute(0, 3, 1, 2)
        x = self.norm(x)
        return x, feat_size


def resample_patch_embed(
        patch_embed: torch.Tensor,
        new_size: List[int],
        interpolation: str = 'bicubic',
        antialias: bool = True,
        verbose: bool = False,
):
    """Resample conv patch-embed weights to a new kernel size.

    Follows the FlexiViT resize-with-pseudoinverse approach of the reference
    (`patch_embed.py:resample_patch_embed`): build the resize matrix from basis
    vectors and apply its pseudo-inverse-transpose so that
    `resized_weight @ resized_patch ≈ weight @ patch`.
    """
    import numpy as np
    assert len(patch_embed.shape) == 4, "Four dimensions expected"
    assert len(new_size) == 2, "New shape should only be hw"
    old_size = patch_embed.shape[-2:]
    if tuple(old_size) == tuple(new_size):
        return patch_embed

    if verbose:
        _logger.info(f"Resize patch embedding {patch_embed.shape} to {new_size}, w/ {interpolation} interpolation.")

    def resize(x_np, _new_size):
        x_tf = torch.Tensor(x_np)[None, None, ...]
        x_upsampled = F.interpolate(
            x_tf, size=_new_size, mode=interpolation, antialias=antialias)[0, 0, ...].numpy()
        return x_upsampled

    def get_resize_mat(_old_size, _new_size):
        mat = []
        for i in range(np.prod(_old_size)):
            basis_vec = np.zeros(_old_size)
            basis_vec[np.unravel_index(i, _old_size)] = 1.
            mat.append(resize(basis_vec, _new_size).reshape(-1))
        return np.stack(mat).T

    resize_mat = get_resize_mat(old_size, new_size)
    resize_mat_pinv = torch.tensor(np.linalg.pinv(resize_mat.T), device=patch_embed.device)

    def resample_kernel(kernel):
        resampled_kernel = resize_mat_pinv @ kernel.reshape(-1)
        return resampled_kernel.reshape(new_size)

    v_resample_kernel = torch.vmap(torch.vmap(resample_kernel, 0, 0), 1, 1)
    orig_dtype = patch_embed.dtype
    patch_embed = patch_embed.float()
    patch_embed = v_resample_kernel(patch_embed)
    return patch_embed.to(orig_dtype)
