"""Test-time pooling: evaluate a classifier at a resolution above its training
size by sliding the original pool window and avg+max pooling the logit map.

Behavioral parity: /root/reference/timm/layers/test_time_pool.py
(TestTimePoolHead wrapping + the larger-than-default-size activation rule).
"""
import logging

import torch.nn.functional as F
from torch import nn

from .adaptive_avgmax_pool import adaptive_avgmax_pool2d

_logger = logging.getLogger(__name__)

__all__ = ['TestTimePoolHead', 'apply_test_time_pool']


class TestTimePoolHead(nn.Module):
    """Replaces a model's pooled classifier with a convolutional one applied
    to the unpooled feature map, then avg+max pools the class logits."""

    def __init__(self, base, original_pool=7):
        super().__init__()
        self.base = base
        self.original_pool = original_pool
        self.fc = self._as_conv_classifier(base)
        base.reset_classifier(0)  # head now lives in self.fc

    @staticmethod
    def _as_conv_classifier(base):
        head = base.get_classifier()
        if isinstance(head, nn.Conv2d):
            return head
        conv = nn.Conv2d(base.num_features, base.num_classes, kernel_size=1, bias=True)
        conv.weight.data.copy_(head.weight.data.view(conv.weight.shape))
        conv.bias.data.copy_(head.bias.data.view(conv.bias.shape))
        return conv

    def forward(self, x):
        feats = self.base.forward_features(x)
        # stride-1 window of the train-time pool size keeps per-position logits
        feats = F.avg_pool2d(feats, kernel_size=self.original_pool, stride=1)
        logits = self.fc(feats)
        return adaptive_avgmax_pool2d(logits, 1).flatten(1)


def apply_test_time_pool(model, config, use_test_size=False):
    """Wrap ``model`` in TestTimePoolHead when the eval input size exceeds the
    pretrained default in both spatial dims.  Returns (model, enabled)."""
    cfg = getattr(model, 'default_cfg', None)
    if not cfg:
        return model, False
    if use_test_size and 'test_input_size' in cfg:
        trained_size = cfg['test_input_size']
    else:
        trained_size = cfg['input_size']
    target = config['input_size']
    if not (target[-1] > trained_size[-1] and target[-2] > trained_size[-2]):
        return model, False
    _logger.info(
        'Target input size %s > pretrained default %s, using test time pooling',
        str(target[-2:]), str(trained_size[-2:]))
    return TestTimePoolHead(model, original_pool=cfg['pool_size']), True
