"""Tile-level flash-attention reference (the blueprint for the CDNA4
attention kernels — see docs/ROADMAP.md item 1).

This mirrors, op for op, the tiling the HIP kernels will use: Q tiles of
BQ rows iterate KV tiles of BK rows with online-softmax rescaling; the
backward recomputes P from the saved per-row lse and accumulates
dq/dk/dv tile by tile. Everything is fp32 math over bf16-roundable
inputs so the GPU kernel can be diff-tested against it tile-for-tile.

Conventions (per batch*head slice):
    q, k, v: [N, D]; causal masking optional.
    fwd returns (o [N, D], lse [N]) with lse = m + log(sumexp) per row.
    bwd consumes (do, q, k, v, o, lse) and returns (dq, dk, dv).
"""

import math
from typing import Optional, Tuple

import torch


def flash_attn_fwd_tiled(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = True,
    bq: int = 32,
    bk: int = 64,
    scale: Optional[float] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    N, D = q.shape
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    o = torch.zeros(N, D, dtype=torch.float32)
    lse = torch.empty(N, dtype=torch.float32)

    for i0 in range(0, N, bq):
        i1 = min(i0 + bq, N)
        qi = q[i0:i1].float()
        m = torch.full((i1 - i0,), -float('inf'))
        s_sum = torch.zeros(i1 - i0)
        acc = torch.zeros(i1 - i0, D)
        kv_end = i1 if causal else N
        for j0 in range(0, kv_end, bk):
            j1 = min(j0 + bk, N)
            s = (qi @ k[j0:j1].float().T) * scale  # [bq, bk]  (QK^T MFMA)
            if causal:
                row = torch.arange(i0, i1).unsqueeze(1)
                col = torch.arange(j0, j1).unsqueeze(0)
                s = s.masked_fill(col > row, -float('inf'))
            m_new = torch.maximum(m, s.max(dim=1).values)
            # rescale previous accumulator and sum (online softmax)
            alpha = torch.where(torch.isinf(m), torch.zeros_like(m), torch.exp(m - m_new))
            p = torch.exp(s - m_new.unsqueeze(1))  # [bq, bk]
            p = torch.nan_to_num(p, nan=0.0)  # -inf - -inf rows
            s_sum = s_sum * alpha + p.sum(dim=1)
            acc = acc * alpha.unsqueeze(1) + p @ v[j0:j1].float()  # (PV MFMA)
            m = m_new
        o[i0:i1] = acc / s_sum.unsqueeze(1)
        lse[i0:i1] = m + torch.log(s_sum)
    return o, lse


def flash_attn_bwd_tiled(
    do: torch.Tensor,
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    o: torch.Tensor,
    lse: torch.Tensor,
    causal: bool = True,
    bq: int = 32,
    bk: int = 64,
    scale: Optional[float] = None,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    N, D = q.shape
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    qf, kf, vf, dof, of = (t.float() for t in (q, k, v, do, o))

    # delta = rowsum(dO * O)  (the bwd_preprocess kernel)
    delta = (dof * of).sum(dim=1)  # [N]

    dq = torch.zeros(N, D)
    dk = torch.zeros(N, D)
    dv = torch.zeros(N, D)

    # kv-tile outer loop (the dk/dv kernel); q-tile inner loop
    for j0 in range(0, N, bk):
        j1 = min(j0 + bk, N)
        i_start = j0 if causal else 0
        for i0 in range(i_start, N, bq):
            i1 = min(i0 + bq, N)
            s = (qf[i0:i1] @ kf[j0:j1].T) * scale
            if causal:
                row = torch.arange(i0, i1).unsqueeze(1)
                col = torch.arange(j0, j1).unsqueeze(0)
                s = s.masked_fill(col > row, -float('inf'))
            p = torch.exp(s - lse[i0:i1].unsqueeze(1))  # [bq, bk], recomputed
            p = torch.nan_to_num(p, nan=0.0)
            dv[j0:j1] += p.T @ dof[i0:i1]  # (P^T dO MFMA)
            dp = dof[i0:i1] @ vf[j0:j1].T  # (dO V^T MFMA)
            ds = p * (dp - delta[i0:i1].unsqueeze(1)) * scale
            dq[i0:i1] += ds @ kf[j0:j1]  # (dS K MFMA)
            dk[j0:j1] += ds.T @ qf[i0:i1]  # (dS^T Q MFMA)
    return dq, dk, dv


def mha_fwd_tiled(q, k, v, causal=True, bq=32, bk=64):
    """[B, H, N, D] wrapper over the per-slice tile algorithm."""
    B, H, N, D = q.shape
    o = torch.empty(B, H, N, D, dtype=torch.float32)
    lse = torch.empty(B, H, N, dtype=torch.float32)
    for b in range(B):
        for h in range(H):
            o[b, h], lse[b, h] = flash_attn_fwd_tiled(q[b, h], k[b, h], v[b, h], causal, bq, bk)
    return o, lse
