"""Rectilinear-grid linear interpolation (reference
`timm/layers/interpolate.py`; used for rel-pos-bias table resizing)."""
from itertools import product

import torch


class RegularGridInterpolator:
    """Multilinear interpolation on a (possibly unevenly spaced) grid;
    matches scipy's RegularGridInterpolator in 'linear' mode."""

    def __init__(self, points, values):
        assert isinstance(points, (tuple, list))
        assert isinstance(values, torch.Tensor)
        self.points = points
        self.values = values
        self.ms = list(values.shape)
        self.n = len(points)
        assert len(self.ms) == self.n
        for i, p in enumerate(points):
            assert isinstance(p, torch.Tensor)
            assert p.shape[0] == values.shape[i]

    def __call__(self, points_to_interp):
        assert len(points_to_interp) == len(self.points)
        K = points_to_interp[0].shape[0]
        assert all(x.shape[0] == K for x in points_to_interp)

        idxs, dists, spans = [], [], []
        for p, x in zip(self.points, points_to_interp):
            right = torch.bucketize(x, p)
            right[right >= p.shape[0]] = p.shape[0] - 1
            left = (right - 1).clamp(0, p.shape[0] - 1)
            d_left = (x - p[left]).clamp_min(0.)
            d_right = (p[right] - x).clamp_min(0.)
            degenerate = (d_left == 0) & (d_right == 0)
            d_left[degenerate] = d_right[degenerate] = 1.
            idxs.append((left, right))
            dists.append((d_left, d_right))
            spans.append(d_left + d_right)

        numerator = 0.
        for corner in product([0, 1], repeat=self.n):
            take = [idx[onoff] for onoff, idx in zip(corner, idxs)]
            weights = [dist[1 - onoff] for onoff, dist in zip(corner, dists)]
            numerator += self.values[tuple(take)] * torch.prod(torch.stack(weights), dim=0)
        return numerator / torch.prod(torch.stack(spans), dim=0)
