"""ImageNet ReaL-labels re-scoring (arxiv 2006.07159).

Behavioral parity: /root/reference/timm/data/real_labels.py:13.  The real.json
multi-label file must be supplied locally (no network in this environment).
"""
import json
import os

import numpy as np

__all__ = ['RealLabelsImagenet']


class RealLabelsImagenet:
    """Accumulates top-k correctness against the ReaL multi-label set, keyed
    by the canonical val filenames (fed in loader order)."""

    def __init__(self, filenames, real_json=None, topk=(1, 5)):
        if real_json is None:
            raise FileNotFoundError('real_json file is required (no network to fetch real.json)')
        with open(real_json) as f:
            label_lists = json.load(f)
        self.real_labels = {
            f'ILSVRC2012_val_{i + 1:08d}.JPEG': labels
            for i, labels in enumerate(label_lists)
        }
        self.filenames = filenames
        assert len(self.filenames) == len(self.real_labels)
        self.topk = topk
        self.is_correct = {k: [] for k in topk}
        self.sample_idx = 0

    def add_result(self, output):
        """Consume one batch of logits (order must match ``filenames``)."""
        _, top_pred = output.topk(max(self.topk), 1, True, True)
        for pred in top_pred.cpu().numpy():
            name = os.path.basename(self.filenames[self.sample_idx])
            valid = self.real_labels[name]
            if valid:  # images with an empty label set are excluded from ReaL
                for k in self.topk:
                    self.is_correct[k].append(any(p in valid for p in pred[:k]))
            self.sample_idx += 1

    def get_accuracy(self, k=None):
        if k is None:
            return {k: float(np.mean(self.is_correct[k])) * 100 for k in self.topk}
        return float(np.mean(self.is_correct[k])) * 100
