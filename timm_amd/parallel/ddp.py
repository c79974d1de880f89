"""Bucketed data-parallel gradient reducer over RCCL/xGMI.

The reference delegates gradient all-reduce to torch DDP's C++ reducer
(`timm/task/classification.py:65`).  Here we own that logic:

 * gradients live in flat per-bucket buffers from the start — `p.grad` is a
   view into the bucket, so backward accumulates directly into the flat
   buffer (no copy pass before the collective)
 * buckets are ordered by reverse parameter order (≈ backward completion
   order) and sized for the xGMI ring: each of the 7 point-to-point links is
   ≈153 GB/s, so buckets must be large enough to amortize per-collective
   launch cost but small enough to overlap with remaining backward —
   default 50 MB (bigger than NV-tuned 25 MB: per-link ring bandwidth is the
   bound, fewer+larger collectives win on this topology)
 * when the last grad of a bucket is accumulated, its all_reduce(SUM) is
   launched async on RCCL's comm stream — overlapping with the rest of
   backward; `finish_gradient_sync()` waits and applies the 1/world scale
 * `no_sync()` skips launching collectives (grad accumulation), matching the
   reference's `task.no_sync()` (`timm/task/task.py:231`)

Works with any torch.distributed backend (RCCL on ROCm GPUs, gloo for the
CPU multi-process tests).
"""
import contextlib
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
from torch import nn

__all__ = ['BucketedDataParallel', 'DistributedDataParallel']


class _Bucket:
    __slots__ = ('params', 'flat', 'views', 'pending', 'work', 'ready_count')

    def __init__(self, params: List[torch.nn.Parameter]):
        self.params = params
        numel = sum(p.numel() for p in params)
        p0 = params[0]
        self.flat = torch.zeros(numel, dtype=p0.dtype, device=p0.device)
        self.views = []
        offset = 0
        for p in params:
            self.views.append(self.flat[offset:offset + p.numel()].view_as(p))
            offset += p.numel()
        self.ready_count = 0
        self.work = None


class BucketedDataParallel(nn.Module):
    """DDP wrapper with in-house bucketed all-reduce overlap."""

    def __init__(
            self,
            module: nn.Module,
            bucket_cap_mb: float = 50.,
            process_group=None,
            broadcast_params: bool = True,
            gradient_as_bucket_view: bool = True,
    ):
        super().__init__()
        assert dist.is_initialized(), 'torch.distributed must be initialized'
        self.module = module
        self.process_group = process_group
        self.world_size = dist.get_world_size(process_group)
        self._require_sync = True
        self._hooks = []

        if broadcast_params:
            with torch.no_grad():
                for t in module.state_dict().values():
                    if isinstance(t, torch.Tensor) and t.numel() > 0 and t.device.type != 'meta':
                        dist.broadcast(t, src=0, group=process_group)

        # build buckets in reverse parameter order (approximates backward order)
        params = [p for p in module.parameters() if p.requires_grad]
        cap = int(bucket_cap_mb * 1024 * 1024)
        buckets: List[List[nn.Parameter]] = []
        cur: List[nn.Parameter] = []
        cur_bytes = 0
        cur_dtype = None
        for p in reversed(params):
            nbytes = p.numel() * p.element_size()
            if cur and (cur_bytes + nbytes > cap or p.dtype != cur_dtype):
                buckets.append(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += nbytes
            cur_dtype = p.dtype
        if cur:
            buckets.append(cur)

        self._buckets = [_Bucket(b) for b in buckets]

        # Dedicated comm stream (GPU backends): each ready bucket's all-reduce
        # AND its 1/world scale run on this stream, overlapping with the rest
        # of backward; finish_gradient_sync() then only inserts a device-side
        # wait on the default stream instead of serializing the scales there.
        self._comm_stream = None
        self._bucket_events = None
        if torch.cuda.is_available() and self._buckets and self._buckets[0].flat.is_cuda:
            self._comm_stream = torch.cuda.Stream()
            self._bucket_events = [torch.cuda.Event() for _ in self._buckets]
        self._param_bucket: Dict[nn.Parameter, tuple] = {}
        for bi, bucket in enumerate(self._buckets):
            for pi, p in enumerate(bucket.params):
                self._param_bucket[p] = (bi, pi)
                # grad lives in the flat buffer
                p.grad = bucket.views[pi]
                hook = p.register_post_accumulate_grad_hook(self._make_hook(bi))
                self._hooks.append(hook)

    def _make_hook(self, bucket_idx: int):
        def hook(param):
            if not self._require_sync:
                return
            bucket = self._buckets[bucket_idx]
            bucket.ready_count += 1
            if bucket.ready_count == len(bucket.params):
                bucket.ready_count = 0
                if self._comm_stream is not None:
                    # hand the bucket to the comm stream: reduce + scale there
                    ready = torch.cuda.Event()
                    ready.record()  # grads for this bucket are complete on the current stream
                    with torch.cuda.stream(self._comm_stream):
                        ready.wait()
                        dist.all_reduce(
                            bucket.flat, op=dist.ReduceOp.SUM, group=self.process_group)
                        bucket.flat.mul_(1.0 / self.world_size)
                        self._bucket_events[bucket_idx].record()
                    bucket.work = True
                else:
                    bucket.work = dist.all_reduce(
                        bucket.flat, op=dist.ReduceOp.SUM, group=self.process_group, async_op=True)
        return hook

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    @contextlib.contextmanager
    def no_sync(self):
        """Disable gradient all-reduce within this context (grad accumulation)."""
        prev = self._require_sync
        self._require_sync = False
        try:
            yield
        finally:
            self._require_sync = prev

    def finish_gradient_sync(self):
        """Wait for in-flight bucket collectives + apply 1/world averaging.

        Call between loss.backward() and optimizer.step().
        """
        if not self._require_sync:
            return
        inv = 1.0 / self.world_size
        for bi, bucket in enumerate(self._buckets):
            if bucket.work is not None:
                if self._comm_stream is not None:
                    # device-side wait: optimizer kernels on the default stream
                    # order after the comm stream's reduce+scale, host never blocks
                    self._bucket_events[bi].wait()
                else:
                    bucket.work.wait()
                    bucket.flat.mul_(inv)
                bucket.work = None
            elif bucket.ready_count:
                # A partially-ready bucket means some params received no grad
                # this backward (unused branch / aux head).  Silently skipping
                # the reduce would let ranks apply different gradients, and
                # force-reducing here can reorder collectives across ranks
                # (NCCL deadlock hazard), so fail loudly like torch DDP does
                # without find_unused_parameters.
                bucket.ready_count = 0
                raise RuntimeError(
                    'BucketedDataParallel: a gradient bucket is only partially ready at '
                    'sync time — some parameters received no gradient in this backward '
                    '(unused model branch?). All parameters must contribute a grad every '
                    'step, or the unused branch must be detached from the reducer.'
                )

    def zero_grad_buckets(self, set_to_none: bool = False):
        """Zero the flat gradient buffers (p.grad views stay attached)."""
        for bucket in self._buckets:
            bucket.flat.zero_()
            bucket.ready_count = 0
            bucket.work = None
        # re-attach views in case an optimizer detached them
        for bucket in self._buckets:
            for p, view in zip(bucket.params, bucket.views):
                if p.grad is not view:
                    p.grad = view

    # -- state dict passthrough so checkpoints keep `module.`-free keys handled
    #    by clean_state_dict, matching reference DDP behaviour --
    def state_dict(self, *args, **kwargs):
        return super().state_dict(*args, **kwargs)


# Alias matching the torch/naming used around the reference
DistributedDataParallel = BucketedDataParallel
