"""Distributed init + manual collectives (reference `timm/utils/distributed.py`).

MI355X-native: backend is RCCL (`"nccl"` on ROCm IS RCCL), 1 process : 1 GPU
over xGMI.  torchrun-compatible env rendezvous (RANK/LOCAL_RANK/WORLD_SIZE/
MASTER_ADDR/MASTER_PORT) plus SLURM/MPI fallbacks, matching the reference's
`init_distributed_device` (`utils/distributed.py:80-173`).
"""
import logging
import os
from typing import Optional

import torch
import torch.distributed as dist

_logger = logging.getLogger(__name__)


def reduce_tensor(tensor: torch.Tensor, n: int) -> torch.Tensor:
    """all_reduce(SUM)/n — scalar loss/metric averaging (reference `:17-21`)."""
    rt = tensor.clone()
    dist.all_reduce(rt, op=dist.ReduceOp.SUM)
    rt /= n
    return rt


def distribute_bn(model: torch.nn.Module, world_size: int, reduce: bool = False):
    """Sync BN running stats across ranks at epoch end (reference `:24-34`)."""
    from .model import unwrap_model
    model = unwrap_model(model)
    for bn_name, bn_buf in model.named_buffers(recurse=True):
        if ('running_mean' in bn_name) or ('running_var' in bn_name):
            if reduce:
                dist.all_reduce(bn_buf, op=dist.ReduceOp.SUM)
                bn_buf /= float(world_size)
            else:
                dist.broadcast(bn_buf, 0)


def is_global_primary(args) -> bool:
    return args.rank == 0


def is_local_primary(args) -> bool:
    return args.local_rank == 0


def is_primary(args, local: bool = False) -> bool:
    return is_local_primary(args) if local else is_global_primary(args)


def is_distributed_env() -> bool:
    if 'WORLD_SIZE' in os.environ:
        return int(os.environ['WORLD_SIZE']) > 1
    if 'SLURM_NTASKS' in os.environ:
        return int(os.environ['SLURM_NTASKS']) > 1
    return False


def world_info_from_env():
    local_rank = 0
    for v in ('LOCAL_RANK', 'MPI_LOCALRANKID', 'SLURM_LOCALID', 'OMPI_COMM_WORLD_LOCAL_RANK'):
        if v in os.environ:
            local_rank = int(os.environ[v])
            break
    global_rank = 0
    for v in ('RANK', 'PMI_RANK', 'SLURM_PROCID', 'OMPI_COMM_WORLD_RANK'):
        if v in os.environ:
            global_rank = int(os.environ[v])
            break
    world_size = 1
    for v in ('WORLD_SIZE', 'PMI_SIZE', 'SLURM_NTASKS', 'OMPI_COMM_WORLD_SIZE'):
        if v in os.environ:
            world_size = int(os.environ[v])
            break
    return local_rank, global_rank, world_size


def init_distributed_device(args):
    """Initialize process group + bind this process to its GPU.

    Populates args.distributed/args.world_size/args.rank/args.local_rank and
    returns the torch.device.
    """
    args.distributed = False
    args.world_size = 1
    args.rank = 0
    args.local_rank = 0
    result = init_distributed_device_so(
        device=getattr(args, 'device', 'cuda'),
        dist_backend=getattr(args, 'dist_backend', None),
        dist_url=getattr(args, 'dist_url', None),
    )
    args.device = result['device']
    args.world_size = result['world_size']
    args.rank = result['global_rank']
    args.local_rank = result['local_rank']
    args.distributed = result['distributed']
    device = torch.device(args.device)
    return device


def init_distributed_device_so(
        device: str = 'cuda',
        dist_backend: Optional[str] = None,
        dist_url: Optional[str] = None,
):
    # Distributed training = training on more than one GPU.
    # Works in both single and multi-node scenarios.
    distributed = False
    world_size = 1
    global_rank = 0
    local_rank = 0
    device_type, *device_idx = device.split(':', maxsplit=1)

    if dist_backend is None:
        # on ROCm the "nccl" backend IS RCCL
        dist_backends = {'xpu': 'ccl', 'hpu': 'hccl', 'cuda': 'nccl', 'npu': 'hccl'}
        dist_backend = dist_backends.get(device_type, 'gloo')
    dist_url = dist_url or 'env://'

    if is_distributed_env():
        if 'SLURM_PROCID' in os.environ and 'RANK' not in os.environ:
            # SLURM without torchrun
            local_rank, global_rank, world_size = world_info_from_env()
            os.environ['LOCAL_RANK'] = str(local_rank)
            os.environ['RANK'] = str(global_rank)
            os.environ['WORLD_SIZE'] = str(world_size)
            dist.init_process_group(backend=dist_backend, init_method=dist_url,
                                    world_size=world_size, rank=global_rank)
        else:
            # DDP via torchrun, torch.distributed.launch
            local_rank, _, _ = world_info_from_env()
            dist.init_process_group(backend=dist_backend, init_method=dist_url)
            world_size = dist.get_world_size()
            global_rank = dist.get_rank()
        distributed = True

    if distributed and device_type == 'cuda' and torch.cuda.is_available():
        device = f'cuda:{local_rank}'
    if device_type == 'cuda' and torch.cuda.is_available():
        if ':' not in device:
            device = f'{device}:0'  # set_device needs an explicit index
        torch.cuda.set_device(device)

    return dict(
        device=device,
        global_rank=global_rank,
        local_rank=local_rank,
        world_size=world_size,
        distributed=distributed,
    )
