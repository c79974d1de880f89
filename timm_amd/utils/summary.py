"""Experiment output dirs + per-epoch metric rows (CSV and optional wandb).

Behavioral parity: /root/reference/timm/utils/summary.py:14,30 (same CSV
column naming: epoch, train_*, eval_*, lr).
"""
import csv
import os

try:
    import wandb
except ImportError:
    wandb = None

__all__ = ['get_outdir', 'update_summary']


def get_outdir(path, *paths, inc=False):
    """mkdir -p the joined path; with ``inc``, suffix -1, -2, ... if taken."""
    base = os.path.join(path, *paths)
    if not os.path.exists(base):
        os.makedirs(base)
        return base
    if not inc:
        return base
    for n in range(1, 100):
        candidate = f'{base}-{n}'
        if not os.path.exists(candidate):
            os.makedirs(candidate)
            return candidate
    raise AssertionError(f'could not find a free increment of {base}')


def update_summary(
        epoch,
        train_metrics,
        eval_metrics,
        filename,
        lr=None,
        write_header=False,
        log_wandb=False,
):
    """Append one epoch row to the summary CSV (and wandb when enabled)."""
    row = {'epoch': epoch}
    row.update({f'train_{k}': v for k, v in train_metrics.items()})
    if eval_metrics:
        row.update({f'eval_{k}': v for k, v in eval_metrics.items()})
    if lr is not None:
        row['lr'] = lr
    if log_wandb and wandb is not None:
        wandb.log(row)
    with open(filename, mode='a') as f:
        writer = csv.DictWriter(f, fieldnames=row.keys())
        if write_header:  # caller tracks first write (resume may append)
            writer.writeheader()
        writer.writerow(row)
