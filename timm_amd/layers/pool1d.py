"""Token-sequence (NLC) global pooling (reference `timm/layers/pool1d.py`)."""
import torch


def global_pool_nlc(
        x: torch.Tensor,
        pool_type: str = 'token',
        num_prefix_tokens: int = 1,
        reduce_include_prefix: bool = False,
):
    """Pool a (B, N, C) sequence: 'token' takes the class token, the rest
    reduce over the non-prefix tokens (or all when reduce_include_prefix)."""
    if not pool_type:
        return x
    if pool_type == 'token':
        return x[:, 0]
    x = x if reduce_include_prefix else x[:, num_prefix_tokens:]
    if pool_type == 'avg':
        return x.mean(dim=1)
    if pool_type == 'avgmax':
        return 0.5 * (x.amax(dim=1) + x.mean(dim=1))
    if pool_type == 'max':
        return x.amax(dim=1)
    raise AssertionError(f'Unknown pool type {pool_type}')
