"""The flash-attention tile blueprint must match plain softmax attention
(and torch autograd) exactly — this is the oracle the round-2 CDNA4
attention kernels will be diff-tested against."""

import math
import sys

import pytest
import torch

from dmlcloud_amd.ops._attention_ref import flash_attn_bwd_tiled, flash_attn_fwd_tiled


def _plain_attention(q, k, v, causal):
    N, D = q.shape
    s = (q @ k.T) / math.sqrt(D)
    if causal:
        mask = torch.triu(torch.ones(N, N, dtype=torch.bool), diagonal=1)
        s = s.masked_fill(mask, -float('inf'))
    p = torch.softmax(s, dim=-1)
    return p @ v


@pytest.mark.parametrize('causal', [True, False])
@pytest.mark.parametrize('n,d,bq,bk', [(128, 64, 32, 64), (96, 32, 32, 32), (130, 64, 32, 64)])
def test_fwd_matches_plain(causal, n, d, bq, bk):
    torch.manual_seed(0)
    q, k, v = (torch.randn(n, d) for _ in range(3))
    o, lse = flash_attn_fwd_tiled(q, k, v, causal=causal, bq=bq, bk=bk)
    ref = _plain_attention(q, k, v, causal)
    torch.testing.assert_close(o, ref, rtol=1e-5, atol=1e-5)
    # lse sanity: softmax denominators reproduce exactly
    s = (q @ k.T) / math.sqrt(d)
    if causal:
        s = s.masked_fill(torch.triu(torch.ones(n, n, dtype=torch.bool), 1), -float('inf'))
    ref_lse = torch.logsumexp(s, dim=-1)
    torch.testing.assert_close(lse, ref_lse, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize('causal', [True, False])
@pytest.mark.parametrize('n,d', [(128, 64), (64, 32)])
def test_bwd_matches_autograd(causal, n, d):
    torch.manual_seed(1)
    q = torch.randn(n, d, requires_grad=True)
    k = torch.randn(n, d, requires_grad=True)
    v = torch.randn(n, d, requires_grad=True)
    out = _plain_attention(q, k, v, causal)
    do = torch.randn(n, d)
    out.backward(do)

    with torch.no_grad():
        o, lse = flash_attn_fwd_tiled(q.detach(), k.detach(), v.detach(), causal=causal)
        dq, dk, dv = flash_attn_bwd_tiled(do, q.detach(), k.detach(), v.detach(), o, lse, causal=causal)

    torch.testing.assert_close(dq, q.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dk, k.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(dv, v.grad, rtol=1e-4, atol=1e-4)


def test_matches_sdpa():
    torch.manual_seed(2)
    n, d = 128, 64
    q, k, v = (torch.randn(1, 1, n, d) for _ in range(3))
    ref = torch.nn.functional.scaled_dot_product_attention(q, k, v, is_causal=True)
    o, _ = flash_attn_fwd_tiled(q[0, 0], k[0, 0], v[0, 0], causal=True)
    torch.testing.assert_close(o, ref[0, 0], rtol=1e-5, atol=1e-5)


if __name__ == '__main__':
    sys.exit(pytest.main([__file__]))
