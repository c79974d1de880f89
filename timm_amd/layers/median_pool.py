"""Median pooling / filtering (reference `timm/layers/median_pool.py`)."""
import torch.nn as nn
import torch.nn.functional as F

from .helpers import to_2tuple, to_4tuple


class MedianPool2d(nn.Module):
    """Median pool (stride=1 makes it a median filter).

    padding is an (l, r, t, b) 4-tuple as taken by F.pad; `same=True`
    overrides it with TF-style same padding for the configured stride.
    """

    def __init__(self, kernel_size=3, stride=1, padding=0, same=False):
        super().__init__()
        self.k = to_2tuple(kernel_size)
        self.stride = to_2tuple(stride)
        self.padding = to_4tuple(padding)
        self.same = same

    def _padding(self, x):
        if not self.same:
            return self.padding
        ih, iw = x.size()[2:]
        ph = max(self.k[0] - (self.stride[0] if ih % self.stride[0] == 0 else ih % self.stride[0]), 0)
        pw = max(self.k[1] - (self.stride[1] if iw % self.stride[1] == 0 else iw % self.stride[1]), 0)
        return (pw // 2, pw - pw // 2, ph // 2, ph - ph // 2)

    def forward(self, x):
        x = F.pad(x, self._padding(x), mode='reflect')
        x = x.unfold(2, self.k[0], self.stride[0]).unfold(3, self.k[1], self.stride[1])
        return x.contiguous().view(x.size()[:4] + (-1,)).median(dim=-1)[0]
