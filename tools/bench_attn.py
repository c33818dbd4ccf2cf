"""Micro-benchmark: experimental MFMA attention forward vs SDPA (AOTriton)."""

import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from dmlcloud_amd import _C

DEV = 'cuda:0'


def time_fn(fn, reps=30, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(reps):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / reps * 1e3  # us


def bench(b, h, n, d=64, causal=True):
    torch.manual_seed(0)
    q = (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16)
    k = (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16)
    v = (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16)
    o = torch.empty_like(q)
    lse = torch.empty(b, h, n, dtype=torch.float32, device=DEV)
    scale = 1.0 / math.sqrt(d)

    t_ours = time_fn(lambda: _C.attn_fwd(q, k, v, o, lse, scale, causal))
    with torch.no_grad():
        t_sdpa = time_fn(
            lambda: torch.nn.functional.scaled_dot_product_attention(q, k, v, is_causal=causal)
        )
    flops = 4 * b * h * n * n * d * (0.5 if causal else 1.0)
    print(
        f'B={b} H={h} N={n} causal={causal}: ours {t_ours:8.1f}us ({flops/t_ours/1e6:6.1f} TF) '
        f'| sdpa {t_sdpa:8.1f}us ({flops/t_sdpa/1e6:6.1f} TF)'
    )


def bench_bwd(b, h, n, d=64, causal=True):
    from dmlcloud_amd.ops.fused_attn import sdpa

    torch.manual_seed(0)
    mk = lambda: (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16).requires_grad_(True)
    q1, k1, v1 = mk(), mk(), mk()
    do = (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16)

    out1 = sdpa(q1, k1, v1, causal=causal)

    def ours_bwd():
        q1.grad = k1.grad = v1.grad = None
        out1.backward(do, retain_graph=True)

    q2, k2, v2 = mk(), mk(), mk()
    out2 = torch.nn.functional.scaled_dot_product_attention(q2, k2, v2, is_causal=causal)

    def sdpa_bwd():
        q2.grad = k2.grad = v2.grad = None
        out2.backward(do, retain_graph=True)

    t_ours = time_fn(ours_bwd, reps=20)
    t_ref = time_fn(sdpa_bwd, reps=20)
    flops = 4 * b * h * n * n * d * (0.5 if causal else 1.0) * 2.5  # bwd ~2.5x fwd
    print(
        f'BWD B={b} H={h} N={n} causal={causal}: ours {t_ours:8.1f}us ({flops/t_ours/1e6:6.1f} TF) '
        f'| sdpa {t_ref:8.1f}us ({flops/t_ref/1e6:6.1f} TF)'
    )


if __name__ == '__main__':
    bench(64, 12, 1024)  # the GPT-2 bench shape
    bench(64, 12, 1024, causal=False)
    bench(16, 16, 2048)
    bench(8, 32, 4096)
    bench_bwd(64, 12, 1024)
    bench_bwd(64, 12, 1024, causal=False)
    bench_bwd(16, 16, 2048)
