"""Space<->depth pixel rearrangement (reference `timm/layers/space_to_depth.py`).

Used by TResNet as an efficient stem: a 4x4 space-to-depth turns the stride-4
stem conv into a dense 1x1 GEMM over 48 input channels.
"""
import torch
import torch.nn as nn


class SpaceToDepth(nn.Module):
    """Fold bs x bs spatial blocks into channels: [N,C,H,W] -> [N, C*bs^2, H/bs, W/bs]."""
    bs: torch.jit.Final[int]

    def __init__(self, block_size: int = 4):
        super().__init__()
        assert block_size == 4
        self.bs = block_size

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        N, C, H, W = x.size()
        x = x.view(N, C, H // self.bs, self.bs, W // self.bs, self.bs)
        x = x.permute(0, 3, 5, 1, 2, 4).contiguous()
        return x.view(N, C * self.bs * self.bs, H // self.bs, W // self.bs)


class DepthToSpace(nn.Module):
    """Inverse of SpaceToDepth: [N,C,H,W] -> [N, C/bs^2, H*bs, W*bs]."""

    def __init__(self, block_size: int):
        super().__init__()
        self.bs = block_size

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        N, C, H, W = x.size()
        x = x.view(N, self.bs, self.bs, C // (self.bs ** 2), H, W)
        x = x.permute(0, 3, 4, 1, 5, 2).contiguous()
        return x.view(N, C // (self.bs ** 2), H * self.bs, W * self.bs)
