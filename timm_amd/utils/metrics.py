"""Streaming eval metrics (reference `timm/utils/metrics.py`)."""

__all__ = ['AverageMeter', 'accuracy']


class AverageMeter:
    """Running mean tracker: update(value, n) accumulates a weighted sum."""

    def __init__(self):
        self.reset()

    def reset(self):
        self.val = 0
        self.sum = 0
        self.count = 0
        self.avg = 0

    def update(self, val, n=1):
        self.val = val
        self.sum += val * n
        self.count += n
        self.avg = self.sum / self.count


def accuracy(output, target, topk=(1,)):
    """Top-k accuracy in percent for each k in topk."""
    maxk = min(max(topk), output.size(1))
    batch = target.size(0)
    pred = output.topk(maxk, dim=1, largest=True, sorted=True).indices.t()
    hits = pred.eq(target.reshape(1, -1).expand_as(pred))
    return [hits[:min(k, maxk)].reshape(-1).float().sum(0) * 100. / batch for k in topk]
