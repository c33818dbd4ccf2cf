"""Autograd integration of the fused bf16 LayerNorm (layernorm.hip).

`LayerNorm` is a drop-in for torch.nn.LayerNorm over the last dimension:
on bf16 device tensors it runs the row-per-wave gfx950 kernels; anything
else falls back to F.layer_norm (which doubles as the test oracle).
"""

import torch
from torch import nn
from torch.nn import functional as F

from . import _C, is_available


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        x = x.contiguous()
        D = x.shape[-1]
        R = x.numel() // D
        y = torch.empty_like(x)
        mean = torch.empty(R, dtype=torch.float32, device=x.device)
        rstd = torch.empty(R, dtype=torch.float32, device=x.device)
        _C.layernorm_fwd(x, weight, bias, y, mean, rstd, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        D = x.shape[-1]
        dx = torch.empty_like(x)
        # per-block dgamma/dbeta partial rows (kLnBwdBlocks in layernorm.hip)
        dgb_ws = torch.empty(256 * 2 * D, dtype=torch.float32, device=x.device)
        dgamma = torch.empty(D, dtype=x.dtype, device=x.device)
        dbeta = torch.empty(D, dtype=x.dtype, device=x.device)
        _C.layernorm_bwd(dy, x, mean, rstd, weight, dx, dgb_ws, dgamma, dbeta)
        return dx, dgamma, dbeta, None


class LayerNorm(nn.Module):
    """LayerNorm over the last dim, fused on gfx950 for bf16 inputs."""

    def __init__(self, normalized_shape, eps: float = 1e-5):
        super().__init__()
        if isinstance(normalized_shape, int):
            normalized_shape = (normalized_shape,)
        assert len(normalized_shape) == 1, 'fused LayerNorm normalizes the last dim only'
        self.normalized_shape = tuple(normalized_shape)
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(normalized_shape))
        self.bias = nn.Parameter(torch.zeros(normalized_shape))

    def forward(self, x):
        D = self.normalized_shape[0]
        if (
            x.is_cuda
            and is_available()
            and x.dtype == torch.bfloat16
            and self.weight.dtype == torch.bfloat16
            and D % 8 == 0
            and D <= 2048
        ):
            return _LayerNormFn.apply(x, self.weight, self.bias, self.eps)
        return F.layer_norm(x, self.normalized_shape, self.weight, self.bias, self.eps)
