"""GCNet global-context block (reference `timm/layers/global_context.py`;
paper arxiv 1904.11492).

A single softmax-attention query pools the feature map into one context
vector, which then modulates the input through scale and/or add MLP
branches (GCA = fuse_add only, used by the *-gc byob models).
"""
import torch.nn.functional as F
from torch import nn

from .create_act import create_act_layer, get_act_layer
from .helpers import make_divisible
from .mlp import ConvMlp
from .norm import LayerNorm2d


class GlobalContext(nn.Module):

    def __init__(
            self,
            channels,
            use_attn=True,
            fuse_add=False,
            fuse_scale=True,
            init_last_zero=False,
            rd_ratio=1. / 8,
            rd_channels=None,
            rd_divisor=1,
            act_layer=nn.ReLU,
            gate_layer='sigmoid',
    ):
        super().__init__()
        act_layer = get_act_layer(act_layer)
        self.conv_attn = nn.Conv2d(channels, 1, kernel_size=1, bias=True) if use_attn else None

        if rd_channels is None:
            rd_channels = make_divisible(channels * rd_ratio, rd_divisor, round_limit=0.)
        mlp = lambda: ConvMlp(channels, rd_channels, act_layer=act_layer, norm_layer=LayerNorm2d)
        self.mlp_add = mlp() if fuse_add else None
        self.mlp_scale = mlp() if fuse_scale else None

        self.gate = create_act_layer(gate_layer)
        self.init_last_zero = init_last_zero
        self.reset_parameters()

    def reset_parameters(self):
        if self.conv_attn is not None:
            nn.init.kaiming_normal_(self.conv_attn.weight, mode='fan_in', nonlinearity='relu')
        if self.mlp_add is not None:
            nn.init.zeros_(self.mlp_add.fc2.weight)

    def forward(self, x):
        B, C, H, W = x.shape

        if self.conv_attn is not None:
            # one softmax query over all positions -> [B, C, 1, 1] context
            weights = F.softmax(self.conv_attn(x).reshape(B, 1, H * W), dim=-1).unsqueeze(3)
            context = (x.reshape(B, C, H * W).unsqueeze(1) @ weights).view(B, C, 1, 1)
        else:
            context = x.mean(dim=(2, 3), keepdim=True)

        if self.mlp_scale is not None:
            x = x * self.gate(self.mlp_scale(context))
        if self.mlp_add is not None:
            x = x + self.mlp_add(context)
        return x
