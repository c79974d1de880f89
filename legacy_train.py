#!/usr/bin/env python3
"""Legacy training entrypoint (reference `legacy_train.py`).

The reference keeps its pre-task-refactor training loop as a separate
script. This framework was built task-based from the start, so the legacy
CLI simply forwards to `train.py` (flag-compatible for the arguments both
accept) after a deprecation notice.
"""
import sys
import warnings

if __name__ == '__main__':
    warnings.warn(
        'legacy_train.py is a compatibility alias; use train.py. '
        'Forwarding all arguments.', DeprecationWarning)
    import train
    train.main()
