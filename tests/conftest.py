import os
import sys

# xdist runs 4 workers on an 8-core box: cap intra-op threads per worker so
# the model matrix doesn't thrash on oversubscribed OpenMP pools
os.environ.setdefault('OMP_NUM_THREADS', '2')

import pytest
import torch

torch.set_num_threads(max(1, (os.cpu_count() or 8) // 4))

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires a (MI355X) GPU")
    config.addinivalue_line("markers", "base: core model tests")
    config.addinivalue_line("markers", "slow: long-running test")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
