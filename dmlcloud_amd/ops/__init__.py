"""Dispatch layer over the gfx950 HIP extension.

Device tensors go through the native kernels (dmlcloud_amd._C, built from
ops/csrc/*.hip for gfx950); CPU tensors use the pure-torch reference
implementations in ops/_reference.py, which double as the differential
test oracle.

Policy: on a GPU box the native extension is REQUIRED — if a device
tensor reaches an op and the extension failed to import, we raise rather
than silently falling back to eager torch.
"""

from typing import List, Optional

import torch

from . import _reference as ref
from ._reference import OP_MAX, OP_MIN, OP_SUM, acc_dtype_for  # noqa: F401

try:
    from dmlcloud_amd import _C  # type: ignore

    _HAS_EXT = True
    _EXT_ERROR = None
except ImportError as e:  # pragma: no cover - exercised only on broken builds
    _C = None
    _HAS_EXT = False
    _EXT_ERROR = e


def is_available() -> bool:
    """True if the native gfx950 extension is importable."""
    return _HAS_EXT


def _require_ext():
    if not _HAS_EXT:
        raise RuntimeError(
            'dmlcloud_amd._C (gfx950 HIP extension) is not available but a device tensor '
            'reached a native op. Build it with `python setup.py build_ext --inplace`. '
            f'Original import error: {_EXT_ERROR}'
        )


# --------------------------------------------------------------------- metrics

# Deterministic two-stage reduction workspace: sized for the maximum grid
# the reduce kernels use (ops_common.h kMaxGrid).
_PARTIALS_CACHE = {}


def _partials_for(device: torch.device, dtype: torch.dtype) -> torch.Tensor:
    key = (device, dtype)
    buf = _PARTIALS_CACHE.get(key)
    if buf is None:
        buf = torch.empty(2048, device=device, dtype=dtype)
        _PARTIALS_CACHE[key] = buf
    return buf


def metric_reduce_into(value: torch.Tensor, acc: torch.Tensor, count: torch.Tensor, op: int):
    """Fully reduce `value`, merge scalar into acc[0], count[0] += 1."""
    if value.is_cuda:
        _require_ext()
        _C.metric_reduce_into(value.contiguous(), acc, count, _partials_for(acc.device, acc.dtype), op)
    else:
        ref.reduce_into_acc(value, acc, count, op)


def metric_accumulate_elementwise(value: torch.Tensor, acc: torch.Tensor, count: torch.Tensor, op: int):
    if value.is_cuda:
        _require_ext()
        _C.metric_accumulate_elementwise(value.contiguous(), acc, count, op)
    else:
        ref.accumulate_elementwise(value, acc, count, op)


def metric_finalize_dims(acc: torch.Tensor, dims: Optional[List[int]], op: int) -> torch.Tensor:
    if acc.is_cuda:
        _require_ext()
        return _C.metric_finalize_dims(acc, list(dims or []), op)
    return ref.finalize_dims(acc, dims, op)


# ----------------------------------------------------------------------- copy

_UNIT_BYTES = 1 << 20


def _build_unit_table(srcs: List[torch.Tensor], dsts: List[torch.Tensor]) -> torch.Tensor:
    """Host-side descriptor table {src_ptr, dst_ptr, nbytes} with <=1MiB units."""
    rows = []
    for s, d in zip(srcs, dsts):
        nbytes = s.numel() * s.element_size()
        assert nbytes == d.numel() * d.element_size(), 'src/dst byte size mismatch'
        sp, dp = s.data_ptr(), d.data_ptr()
        off = 0
        while off < nbytes:
            n = min(_UNIT_BYTES, nbytes - off)
            rows.append((sp + off, dp + off, n))
            off += n
    table = torch.tensor(rows, dtype=torch.int64)
    return table


def chunked_copy(srcs: List[torch.Tensor], dsts: List[torch.Tensor]):
    """Copy each contiguous src into its same-sized contiguous dst.

    On device: a single descriptor-table kernel (copy.hip). On CPU:
    plain torch copies.
    """
    if not srcs:
        return
    if srcs[0].is_cuda:
        _require_ext()
        for t in srcs + dsts:
            assert t.is_contiguous(), 'chunked_copy requires contiguous tensors'
        table = _build_unit_table(srcs, dsts).to(srcs[0].device, non_blocking=True)
        _C.chunked_copy(table, table.shape[0])
    else:
        ref.chunked_copy(srcs, dsts)


# ------------------------------------------------------------------ optimizers


def fused_adam(param, grad, exp_avg, exp_avg_sq, step_t, lr, beta1, beta2, eps, weight_decay, grad_scale=1.0):
    if param.is_cuda:
        _require_ext()
        _C.fused_adam(param, grad, exp_avg, exp_avg_sq, step_t, lr, beta1, beta2, eps, weight_decay, grad_scale)
    else:
        ref.fused_adam_step(param, grad, exp_avg, exp_avg_sq, step_t, lr, beta1, beta2, eps, weight_decay, grad_scale)


def fused_adam_bf16(param, grad, master, exp_avg, exp_avg_sq, step_t, lr, beta1, beta2, eps, weight_decay, grad_scale=1.0):
    if param.is_cuda:
        _require_ext()
        _C.fused_adam_bf16(param, grad, master, exp_avg, exp_avg_sq, step_t, lr, beta1, beta2, eps, weight_decay, grad_scale)
    else:
        ref.fused_adam_bf16_step(
            param, grad, master, exp_avg, exp_avg_sq, step_t, lr, beta1, beta2, eps, weight_decay, grad_scale
        )


def fused_sgd_bf16(param, grad, master, momentum_buf, lr, momentum, weight_decay, grad_scale=1.0):
    use_momentum = momentum_buf is not None and momentum != 0
    if param.is_cuda:
        _require_ext()
        _C.fused_sgd_bf16(
            param,
            grad,
            master,
            momentum_buf if momentum_buf is not None else torch.Tensor(),
            lr,
            momentum,
            weight_decay,
            grad_scale,
            use_momentum,
        )
    else:
        ref.fused_sgd_bf16_step(param, grad, master, momentum_buf if use_momentum else None, lr, momentum, weight_decay, grad_scale)


def fused_sgd(param, grad, momentum_buf, lr, momentum, weight_decay, grad_scale=1.0):
    use_momentum = momentum_buf is not None and momentum != 0
    if param.is_cuda:
        _require_ext()
        _C.fused_sgd(
            param,
            grad,
            momentum_buf if momentum_buf is not None else torch.Tensor(),
            lr,
            momentum,
            weight_decay,
            grad_scale,
            use_momentum,
        )
    else:
        ref.fused_sgd_step(param, grad, momentum_buf if use_momentum else None, lr, momentum, weight_decay, grad_scale)


_NORM_WS_CACHE = {}


def _norm_ws(device):
    ws = _NORM_WS_CACHE.get(device)
    if ws is None:
        ws = (torch.empty(2048, device=device, dtype=torch.float64), torch.empty(2, device=device, dtype=torch.float32))
        _NORM_WS_CACHE[device] = ws
    return ws


def l2_norm(flat: torch.Tensor, norm_scale: float = 1.0) -> torch.Tensor:
    """Deterministic L2 norm of a flat fp32/bf16 buffer; returns fp32[1]
    on device. The result is a view of a per-device scratch buffer
    (stable pointers for hipGraph capture): clone it if you need the
    value to survive a later l2_norm/clip_grad_norm_ call.

    norm_scale pre-multiplies the norm: pass 1/world_size when `flat`
    holds a rank-summed gradient to get the averaged-gradient norm."""
    if flat.is_cuda:
        _require_ext()
        partials, out = _norm_ws(flat.device)
        _C.l2_norm_and_scale(flat, partials, out, -1.0, False, float(norm_scale))
        return out[:1]
    return ref.l2_norm(flat) * norm_scale


def clip_grad_norm_(flat: torch.Tensor, max_norm: float, norm_scale: float = 1.0) -> torch.Tensor:
    """Clip flat grads by global L2 norm in-place; returns norm (device, no sync).

    The clip threshold compares `||flat|| * norm_scale` against max_norm,
    so a flat buffer holding the SUM of per-rank gradients clips with DDP
    semantics when norm_scale = 1/world_size (the later 1/world averaging
    in the fused optimizer then lands exactly on the clipped average)."""
    if flat.is_cuda:
        _require_ext()
        partials, out = _norm_ws(flat.device)
        _C.l2_norm_and_scale(flat, partials, out, float(max_norm), True, float(norm_scale))
        return out[:1]
    norm = ref.l2_norm(flat) * norm_scale
    ref.clip_by_norm_(flat, norm[0], max_norm)
    return norm
