"""Linear w/ safe autocast behaviour (reference `timm/layers/linear.py`)."""
import torch
import torch.nn.functional as F
from torch import nn as nn


class Linear(nn.Linear):
    """Applies a linear transformation to the incoming data.

    Wrapper retained for API parity; on ROCm the GEMM runs through hipBLASLt.
    """

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        if torch.jit.is_scripting():
            bias = self.bias.to(dtype=input.dtype) if self.bias is not None else None
            return F.linear(input, self.weight.to(dtype=input.dtype), bias=bias)
        else:
            return F.linear(input, self.weight, self.bias)
