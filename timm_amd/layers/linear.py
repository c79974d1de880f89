"""Linear with torchscript-safe autocast casting (reference
`timm/layers/linear.py`). The GEMM itself runs through hipBLASLt on ROCm."""
import torch
import torch.nn.functional as F
from torch import nn


class Linear(nn.Linear):
    def forward(self, input: torch.Tensor) -> torch.Tensor:
        if torch.jit.is_scripting():
            # scripted graphs don't see autocast: cast params to match input
            bias = self.bias.to(dtype=input.dtype) if self.bias is not None else None
            return F.linear(input, self.weight.to(dtype=input.dtype), bias=bias)
        return F.linear(input, self.weight, self.bias)
