from .ddp import BucketedDataParallel, DistributedDataParallel
