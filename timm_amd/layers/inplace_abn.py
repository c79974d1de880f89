"""Activated BatchNorm module (reference `timm/layers/inplace_abn.py`).

The reference requires the external `inplace_abn` CUDA package; this
framework computes the same result with standard batch_norm + activation
(the HIP allocator and bf16 activations make the memory saving of the
in-place trick marginal on 288 GB parts). State-dict layout matches the
reference (weight/bias/running_mean/running_var).
"""
import torch
import torch.nn.functional as F
from torch import nn

has_iabn = False  # parity flag: the external in-place package is never used


class InplaceAbn(nn.Module):
    def __init__(
            self,
            num_features,
            eps=1e-5,
            momentum=0.1,
            affine=True,
            apply_act=True,
            act_layer='leaky_relu',
            act_param=0.01,
            drop_layer=None,
    ):
        super().__init__()
        self.num_features = num_features
        self.affine = affine
        self.eps = eps
        self.momentum = momentum
        if apply_act:
            if isinstance(act_layer, str):
                assert act_layer in ('leaky_relu', 'elu', 'identity', '')
                self.act_name = act_layer or 'identity'
            elif act_layer == nn.ELU:
                self.act_name = 'elu'
            elif act_layer == nn.LeakyReLU:
                self.act_name = 'leaky_relu'
            elif act_layer is None or act_layer == nn.Identity:
                self.act_name = 'identity'
            else:
                raise AssertionError(f'Invalid act layer {act_layer} for IABN')
        else:
            self.act_name = 'identity'
        self.act_param = act_param
        if affine:
            self.weight = nn.Parameter(torch.ones(num_features))
            self.bias = nn.Parameter(torch.zeros(num_features))
        else:
            self.register_parameter('weight', None)
            self.register_parameter('bias', None)
        self.register_buffer('running_mean', torch.zeros(num_features))
        self.register_buffer('running_var', torch.ones(num_features))
        self.reset_parameters()

    def reset_parameters(self):
        nn.init.constant_(self.running_mean, 0)
        nn.init.constant_(self.running_var, 1)
        if self.affine:
            nn.init.constant_(self.weight, 1)
            nn.init.constant_(self.bias, 0)

    def forward(self, x):
        x = F.batch_norm(
            x, self.running_mean, self.running_var, self.weight, self.bias,
            self.training, self.momentum, self.eps)
        if self.act_name == 'leaky_relu':
            x = F.leaky_relu(x, negative_slope=self.act_param, inplace=True)
        elif self.act_name == 'elu':
            x = F.elu(x, alpha=self.act_param, inplace=True)
        return x
