"""Autograd wrapper for the MFMA flash-attention kernels.

`sdpa(q, k, v, causal=True)` matches
F.scaled_dot_product_attention(..., is_causal=causal) for bf16
[B, H, N, 64] self-attention device tensors; anything else (other head
dims, cross-attention geometry, non-bf16) falls back to torch SDPA
(AOTriton). The fused path is ON by default whenever usable — set
DMLCLOUD_DISABLE_FUSED_ATTN=1 to force the AOTriton path (the A/B
ladder in profiles/ uses this switch). Numerics are verified in
tests/test_gpu.py.
"""

import math
import os

import torch
from torch.nn import functional as F

from . import _C, is_available


class _SdpaFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, scale):
        B, H, N, D = q.shape
        # empty_like would inherit a transpose-view's strides; o is ours
        o = torch.empty(B, H, N, D, dtype=q.dtype, device=q.device)
        lse = torch.empty(B, H, N, dtype=torch.float32, device=q.device)
        _C.attn_fwd(q, k, v, o, lse, scale, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, dout):
        q, k, v, o, lse = ctx.saved_tensors
        B, H, N, D = q.shape
        if dout.stride(-1) != 1:
            dout = dout.contiguous()
        dq = torch.empty(B, H, N, D, dtype=q.dtype, device=q.device)
        dk = torch.empty_like(dq)
        dv = torch.empty_like(dq)
        delta = torch.empty(B * H * N, dtype=torch.float32, device=q.device)
        _C.attn_bwd(q, k, v, dout, o, lse, dq, dk, dv, delta, ctx.scale, ctx.causal)
        return dq, dk, dv, None, None


def _usable(q, k, v):
    if os.environ.get('DMLCLOUD_DISABLE_FUSED_ATTN', '0') not in ('0', '', 'false'):
        return False
    # The kernels assume self-attention geometry: k/v must match q exactly
    # (cross-attention with a different kv length/head count falls back).
    return (
        q.is_cuda
        and is_available()
        and q.dtype == torch.bfloat16
        and q.dim() == 4
        and q.shape[-1] == 64
        and q.shape[-2] % 64 == 0
        and k.shape == q.shape
        and v.shape == q.shape
        and k.dtype == q.dtype
        and v.dtype == q.dtype
        and k.device == q.device
        and v.device == q.device
    )


def _lastdim_ok(t):
    return t.stride(-1) == 1


def sdpa(q, k, v, causal: bool = True):
    """Scaled dot-product attention; fused MFMA kernels when usable.

    The kernels take per-tensor (batch, head, row) strides, so transpose
    views (e.g. [B,T,H,D] memory viewed as [B,H,T,D]) pass through
    without materialization — only a non-unit last-dim stride forces a
    copy."""
    if _usable(q, k, v):
        scale = 1.0 / math.sqrt(q.shape[-1])
        q = q if _lastdim_ok(q) else q.contiguous()
        k = k if _lastdim_ok(k) else k.contiguous()
        v = v if _lastdim_ok(v) else v.contiguous()
        return _SdpaFn.apply(q, k, v, causal, scale)
    return F.scaled_dot_product_attention(q, k, v, is_causal=causal)


def fused_attention_enabled() -> bool:
    """True unless DMLCLOUD_DISABLE_FUSED_ATTN opts out (the fused path
    is on by default for usable shapes)."""
    return os.environ.get('DMLCLOUD_DISABLE_FUSED_ATTN', '0') in ('0', '', 'false')
