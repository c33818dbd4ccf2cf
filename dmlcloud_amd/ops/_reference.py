"""Pure-torch reference implementations of every native op.

These serve two purposes:
1. CPU execution path (tests run on CPU with the gloo backend).
2. Differential testing oracle for the HIP kernels (tests compare the
   gfx950 kernels against these at fp32/fp64).

Semantics mirror the reference's metric math (reference
dmlcloud/metrics.py:24-41,107-119) but restructured around O(1)
device-resident accumulators instead of a growing list of per-batch
CPU copies.
"""

from typing import List, Optional

import torch

# Reduction op codes shared with the HIP extension (keep in sync with csrc/ops.h)
OP_SUM = 0
OP_MIN = 1
OP_MAX = 2


def acc_dtype_for(dtype: torch.dtype) -> torch.dtype:
    """Accumulator dtype: fp64 for floats (better than the reference's
    fp32 stack+mean), int64 for integral, passthrough for fp64/complex."""
    if dtype.is_floating_point:
        return torch.float64
    if dtype in (torch.int8, torch.uint8, torch.int16, torch.int32, torch.int64, torch.bool):
        return torch.int64
    raise ValueError(f'Unsupported metric dtype {dtype}')


def reduce_into_acc(value: torch.Tensor, acc: torch.Tensor, count: torch.Tensor, op: int):
    """Fully reduce `value` and merge the scalar into acc[0]; count[0] += 1."""
    v = value.to(acc.dtype)
    if op == OP_SUM:
        acc[0] = acc[0] + v.sum()
    elif op == OP_MIN:
        acc[0] = torch.minimum(acc[0], v.min())
    elif op == OP_MAX:
        acc[0] = torch.maximum(acc[0], v.max())
    else:
        raise ValueError(f'Unknown op {op}')
    count[0] = count[0] + 1


def accumulate_elementwise(value: torch.Tensor, acc: torch.Tensor, count: torch.Tensor, op: int):
    """Elementwise-merge `value` into same-shaped acc; count[0] += 1."""
    v = value.to(acc.dtype)
    if op == OP_SUM:
        acc.add_(v)
    elif op == OP_MIN:
        torch.minimum(acc, v, out=acc)
    elif op == OP_MAX:
        torch.maximum(acc, v, out=acc)
    else:
        raise ValueError(f'Unknown op {op}')
    count[0] = count[0] + 1


def finalize_dims(acc: torch.Tensor, dims: Optional[List[int]], op: int) -> torch.Tensor:
    """Reduce accumulator over `dims` (list of dims of the original value)."""
    if dims is None or len(dims) == 0:
        return acc.clone()
    if op == OP_SUM:
        return acc.sum(dim=dims)
    if op == OP_MIN:
        return acc.amin(dim=dims)
    if op == OP_MAX:
        return acc.amax(dim=dims)
    raise ValueError(f'Unknown op {op}')


def chunked_copy(srcs: List[torch.Tensor], dsts: List[torch.Tensor]):
    """Byte-copy each flat src into its flat dst (pack/unpack/interleave core)."""
    for s, d in zip(srcs, dsts):
        d.copy_(s.reshape(d.shape))


def fused_adam_step(
    param: torch.Tensor,
    grad: torch.Tensor,
    exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor,
    step_t: torch.Tensor,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    grad_scale: float,
):
    """Adam on flat fp32 buffers. step_t is a device int32[1] incremented here."""
    step_t += 1
    # device-side bias correction (works under graph capture on GPU)
    t = step_t.to(torch.float32)
    bc1 = 1 - beta1**t
    bc2 = 1 - beta2**t
    g = grad * grad_scale
    if weight_decay != 0:
        g = g + weight_decay * param
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    denom = (exp_avg_sq / bc2).sqrt_().add_(eps)
    param.addcdiv_(exp_avg / bc1, denom, value=-lr)


def fused_adam_bf16_step(
    param: torch.Tensor,
    grad: torch.Tensor,
    master: torch.Tensor,
    exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor,
    step_t: torch.Tensor,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    grad_scale: float,
):
    """Mixed-precision Adam: bf16 params/grads, fp32 master + moments."""
    step_t += 1
    t = step_t.to(torch.float32)
    bc1 = 1 - beta1**t
    bc2 = 1 - beta2**t
    g = grad.to(torch.float32) * grad_scale
    if weight_decay != 0:
        g = g + weight_decay * master
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    denom = (exp_avg_sq / bc2).sqrt_().add_(eps)
    master.addcdiv_(exp_avg / bc1, denom, value=-lr)
    param.copy_(master.to(torch.bfloat16))


def fused_sgd_bf16_step(
    param: torch.Tensor,
    grad: torch.Tensor,
    master: torch.Tensor,
    momentum_buf: Optional[torch.Tensor],
    lr: float,
    momentum: float,
    weight_decay: float,
    grad_scale: float,
):
    g = grad.to(torch.float32) * grad_scale
    if weight_decay != 0:
        g = g + weight_decay * master
    if momentum_buf is not None and momentum != 0:
        momentum_buf.mul_(momentum).add_(g)
        g = momentum_buf
    master.add_(g, alpha=-lr)
    param.copy_(master.to(torch.bfloat16))


def fused_sgd_step(
    param: torch.Tensor,
    grad: torch.Tensor,
    momentum_buf: Optional[torch.Tensor],
    lr: float,
    momentum: float,
    weight_decay: float,
    grad_scale: float,
):
    g = grad * grad_scale
    if weight_decay != 0:
        g = g + weight_decay * param
    if momentum_buf is not None and momentum != 0:
        momentum_buf.mul_(momentum).add_(g)
        g = momentum_buf
    param.add_(g, alpha=-lr)


def l2_norm(flat: torch.Tensor) -> torch.Tensor:
    return flat.to(torch.float32).norm(2).reshape(1)


def clip_by_norm_(flat: torch.Tensor, total_norm: torch.Tensor, max_norm: float):
    """Scale flat in-place by min(1, max_norm / (norm + 1e-6)); device-side."""
    scale = (max_norm / (total_norm + 1e-6)).clamp(max=1.0)
    flat.mul_(scale.to(flat.dtype))
