"""Shared epoch counter usable across DataLoader worker processes
(reference `readers/shared_count.py`)."""
from multiprocessing import Value


class SharedCount:
    """Integer in shared memory; streaming readers read it per-iteration to
    learn the current epoch for deterministic shard shuffling."""

    def __init__(self, epoch: int = 0):
        self.shared_epoch = Value('i', epoch)

    @property
    def value(self) -> int:
        return self.shared_epoch.value

    @value.setter
    def value(self, epoch: int):
        self.shared_epoch.value = epoch
