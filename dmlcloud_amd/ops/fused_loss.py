"""Fused softmax-cross-entropy over large vocabularies (loss.hip).

`cross_entropy(logits, targets)` matches F.cross_entropy(reduction='mean')
for 2D bf16 logits on device (single online max+sum pass forward, single
elementwise pass backward); other inputs fall back to torch.
"""

import torch
from torch.nn import functional as F

from . import _C, is_available


class _CrossEntropyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets):
        logits = logits.contiguous()
        V = logits.shape[-1]
        R = logits.numel() // V
        loss = torch.empty(R, dtype=torch.float32, device=logits.device)
        lse = torch.empty(R, dtype=torch.float32, device=logits.device)
        _C.ce_fwd(logits, targets, loss, lse)
        ctx.save_for_backward(logits, targets, lse)
        return loss.mean()

    @staticmethod
    def backward(ctx, grad_out):
        logits, targets, lse = ctx.saved_tensors
        V = logits.shape[-1]
        R = logits.numel() // V
        # device-side scale: upstream grad / R (graph-capture friendly)
        scale = (grad_out.to(torch.float32) / R).reshape(1).contiguous()
        dlogits = torch.empty_like(logits)
        _C.ce_bwd(logits, targets, lse, scale, dlogits)
        return dlogits, None


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
    """Mean cross-entropy. Fused on gfx950 for 2D bf16 logits."""
    if logits.is_cuda and is_available() and logits.dtype == torch.bfloat16 and logits.dim() == 2:
        return _CrossEntropyFn.apply(logits, targets.contiguous())
    return F.cross_entropy(logits, targets)
