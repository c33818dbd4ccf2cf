"""Fused softmax-cross-entropy over large vocabularies (loss.hip).

`cross_entropy(logits, targets)` matches
F.cross_entropy(reduction='mean', ignore_index=...) for 2D bf16 logits
on device (single online max+sum pass forward, single elementwise pass
backward); other inputs fall back to torch.

ignore_index semantics follow torch: ignored rows contribute no loss and
no gradient, and the mean divides by the number of NON-ignored rows (the
valid count is computed on device — no host sync, graph-capture safe).
A target outside [0, vocab) that is not ignore_index produces NaN loss
(loud) instead of an out-of-bounds read.
"""

import torch
from torch.nn import functional as F

from . import _C, is_available


class _CrossEntropyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        logits = logits.contiguous()
        V = logits.shape[-1]
        R = logits.numel() // V
        loss = torch.empty(R, dtype=torch.float32, device=logits.device)
        lse = torch.empty(R, dtype=torch.float32, device=logits.device)
        _C.ce_fwd(logits, targets, loss, lse, ignore_index)
        # device-side valid count: mean divides by non-ignored rows only
        n_valid = (targets != ignore_index).sum().to(torch.float32)
        ctx.save_for_backward(logits, targets, lse, n_valid)
        ctx.ignore_index = ignore_index
        return loss.sum() / n_valid

    @staticmethod
    def backward(ctx, grad_out):
        logits, targets, lse, n_valid = ctx.saved_tensors
        # device-side scale: upstream grad / n_valid (graph-capture friendly)
        scale = (grad_out.to(torch.float32) / n_valid).reshape(1).contiguous()
        dlogits = torch.empty_like(logits)
        _C.ce_bwd(logits, targets, lse, scale, dlogits, ctx.ignore_index)
        return dlogits, None, None


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor, ignore_index: int = -100) -> torch.Tensor:
    """Mean cross-entropy. Fused on gfx950 for 2D bf16 logits."""
    if logits.is_cuda and is_available() and logits.dtype == torch.bfloat16 and logits.dim() == 2:
        return _CrossEntropyFn.apply(logits, targets.contiguous(), ignore_index)
    return F.cross_entropy(logits, targets, ignore_index=ignore_index)
