"""Filter Response Normalization (reference `timm/layers/filter_response_norm.py`).

Paper: Filter Response Normalization Layer — https://arxiv.org/abs/1911.09737
Per-channel instance RMS normalization + affine, with optional TLU
(thresholded linear unit) activation.
"""
from typing import Optional, Type

import torch
import torch.nn as nn

from .create_act import create_act_layer


def inv_instance_rms(x: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    rms = x.square().float().mean(dim=(2, 3), keepdim=True).add(eps).rsqrt().to(x.dtype)
    return rms.expand(x.shape)


class FilterResponseNormTlu2d(nn.Module):
    def __init__(
            self,
            num_features: int,
            apply_act: bool = True,
            eps: float = 1e-5,
            rms: bool = True,
            **_,
    ):
        super().__init__()
        self.apply_act = apply_act
        self.rms = rms
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.tau = nn.Parameter(torch.zeros(num_features)) if apply_act else None

    def reset_parameters(self):
        nn.init.ones_(self.weight)
        nn.init.zeros_(self.bias)
        if self.tau is not None:
            nn.init.zeros_(self.tau)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        assert x.dim() == 4, 'expected 4D input'
        x_dtype = x.dtype
        v_shape = (1, -1, 1, 1)
        x = x * inv_instance_rms(x, self.eps)
        x = x * self.weight.view(v_shape).to(dtype=x_dtype) + self.bias.view(v_shape).to(dtype=x_dtype)
        return torch.maximum(x, self.tau.reshape(v_shape).to(dtype=x_dtype)) if self.tau is not None else x


class FilterResponseNormAct2d(nn.Module):
    def __init__(
            self,
            num_features: int,
            apply_act: bool = True,
            act_layer: Type[nn.Module] = nn.ReLU,
            inplace: Optional[bool] = None,
            rms: bool = True,
            eps: float = 1e-5,
            **_,
    ):
        super().__init__()
        if act_layer is not None and apply_act:
            self.act = create_act_layer(act_layer, inplace=inplace)
        else:
            self.act = nn.Identity()
        self.rms = rms
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))

    def reset_parameters(self):
        nn.init.ones_(self.weight)
        nn.init.zeros_(self.bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        assert x.dim() == 4, 'expected 4D input'
        x_dtype = x.dtype
        v_shape = (1, -1, 1, 1)
        x = x * inv_instance_rms(x, self.eps)
        x = x * self.weight.view(v_shape).to(dtype=x_dtype) + self.bias.view(v_shape).to(dtype=x_dtype)
        return self.act(x)
