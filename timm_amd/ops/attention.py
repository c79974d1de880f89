"""Fused multi-head attention dispatch (SDPA replacement).

The reference leans on `F.scaled_dot_product_attention`
(`timm/layers/attention.py:124-129`, `timm/models/eva.py:246-251`).  Here the
device path is a hand-written flash-style gfx950 HIP kernel:

 * forward: MFMA 16x16x32 bf16 QK^T + online softmax + PV, returns O and
   logsumexp (LSE) per row.  Supports optional additive mask (NaFlex padding
   masks, rel-pos bias broadcast over batch).
 * backward: exact recompute using hipBLASLt GEMMs (torch.matmul) with the
   saved LSE — the GEMM-shaped work of the backward runs on the MFMA library
   path; a fully-fused bwd kernel is a later optimization.

CPU path = reference math composition (fp32 softmax), which doubles as the
numerics oracle for the GPU tests.
"""
import math
from typing import Optional

import torch
import torch.nn.functional as F

from . import _load_extension


def attention_available(q: torch.Tensor) -> bool:
    if not q.is_cuda or _load_extension() is None:
        return False
    D = q.shape[-1]
    return D <= 128 and D % 32 == 0 and q.dtype == torch.bfloat16


def _math_sdpa(q, k, v, attn_mask=None, scale=None):
    """Reference composition (matches F.scaled_dot_product_attention semantics)."""
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    attn = (q.float() @ k.float().transpose(-2, -1)) * scale
    if attn_mask is not None:
        attn = attn + attn_mask.float()
    attn = attn.softmax(dim=-1)
    out = attn @ v.float()
    return out.to(q.dtype)


class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, attn_mask, scale):
        ext = _load_extension()
        # kernel reads strided q/k/v (e.g. views into the packed qkv tensor)
        # as long as head_dim is contiguous and 16B-alignment holds
        def _ok(t):
            return t.stride(-1) == 1 and all(s % 8 == 0 or s == 0 for s in t.stride()[:3])
        if not _ok(q):
            q = q.contiguous()
        if not _ok(k):
            k = k.contiguous()
        if not _ok(v):
            v = v.contiguous()
        if attn_mask is not None:
            attn_mask = attn_mask.contiguous().to(torch.float32)
        o, lse = ext.attention_fwd(q, k, v, attn_mask, scale)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.attn_mask = attn_mask
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        ext = _load_extension()
        scale = ctx.scale
        # exact flash backward: recompute P from saved LSE. GEMMs stay bf16
        # (MFMA via hipBLASLt, fp32 internal accum); the softmax-recompute
        # elementwise runs in fused HIP kernels (attn_bwd_preprocess/_softmax/_ds).
        if do.dim() == 4 and do.stride(-1) == 1 and do.shape[-1] % 32 == 0 and do.shape[-1] <= 128:
            # fused: contiguous dO + delta = rowsum(dO*O) in one pass
            do, delta = ext.attn_bwd_preprocess(do, o)
        else:
            do = do.contiguous()
            delta = (do * o).float().sum(-1)  # [B,H,Nq] fp32
        s = (q @ k.transpose(-2, -1)).contiguous()  # bf16 GEMM
        p = ext.attn_bwd_softmax(s, lse, ctx.attn_mask, scale)  # exp(s*scale+mask-lse)
        dv = p.transpose(-2, -1) @ do
        dp = (do @ v.transpose(-2, -1)).contiguous()  # bf16 GEMM
        ds = ext.attn_bwd_ds(p, dp, delta, scale)
        dq = ds @ k
        dk = ds.transpose(-2, -1) @ q
        return dq, dk, dv, None, None


def flash_attention(
        q: torch.Tensor,
        k: torch.Tensor,
        v: torch.Tensor,
        attn_mask: Optional[torch.Tensor] = None,
        dropout_p: float = 0.,
        scale: Optional[float] = None,
) -> torch.Tensor:
    """Fused attention. q,k,v: [B, H, N, D]. attn_mask: additive, broadcastable to [B,H,Nq,Nk]."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if dropout_p > 0.:
        # attention-dropout path falls back to composition (rare in configs we target)
        return _dropout_sdpa(q, k, v, attn_mask, dropout_p, scale)
    if attention_available(q):
        if attn_mask is not None:
            # kernel accepts [B|1, H|1, Nq, Nk]: keep batch/head dims unexpanded
            # (rel-pos bias is [1,H,N,N], padding masks [B,1,N,N]) so the
            # contiguous copy below stays small.
            while attn_mask.dim() < 4:
                attn_mask = attn_mask.unsqueeze(0)
            attn_mask = attn_mask.expand(
                attn_mask.shape[0], attn_mask.shape[1], q.shape[2], k.shape[2])
        return _FlashAttnFn.apply(q, k, v, attn_mask, scale)
    if q.is_cuda:
        from . import use_hip
        if use_hip(q):
            # ext present but shape/dtype unsupported (e.g. fp32, D>128): exact composition
            return _math_sdpa(q, k, v, attn_mask, scale)
    return _math_sdpa(q, k, v, attn_mask, scale)


def _dropout_sdpa(q, k, v, attn_mask, dropout_p, scale):
    attn = (q @ k.transpose(-2, -1)) * scale
    if attn_mask is not None:
        attn = attn + attn_mask
    attn = attn.softmax(dim=-1)
    attn = F.dropout(attn, p=dropout_p, training=True)
    return attn @ v
