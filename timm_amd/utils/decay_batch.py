"""Batch-size decay for OOM retry (reference `timm/utils/decay_batch.py:8,32`)."""


def decay_batch_step(batch_size, num_intra_steps=2, no_odd=False):
    """Decay batch size in steps w/ intermediate fractions between powers of two."""
    if batch_size <= 1:
        return 0
    base_batch_size = int(2 ** (math_log2_floor(batch_size)))
    step = max(base_batch_size // num_intra_steps, 1)
    batch_size = base_batch_size if base_batch_size < batch_size else batch_size - step
    if no_odd and batch_size % 2:
        batch_size -= 1
    return max(0, batch_size)


def math_log2_floor(x):
    import math
    return math.floor(math.log(x, 2))


def check_batch_size_retry(error_str):
    """Check failure error string to verify conditions for batch-size retry."""
    error_str = error_str.lower()
    if 'required rank' in error_str:
        # Errors involving phrase 'required rank' typically happen when a conv is used that's
        # not compatible with channels_last memory format.
        return False
    if 'illegal' in error_str:
        # 'Illegal memory access' errors in CUDA may leave process in unusable state, best to skip
        return False
    return True
