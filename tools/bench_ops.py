"""Micro-benchmarks of the core gfx950 ops: chunked copy bandwidth,
metric reduction, fused Adam. Run on a GPU box."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from dmlcloud_amd import ops

DEV = 'cuda:0'


def time_fn(fn, reps=50, warmup=10):
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


def bench_copy():
    print('== chunked_copy (pack) bandwidth ==')
    for total_mb, ntensors in [(64, 16), (512, 64), (2048, 128)]:
        per = total_mb * (1 << 20) // ntensors // 4
        srcs = [torch.randn(per, device=DEV) for _ in range(ntensors)]
        flat = torch.zeros(per * ntensors, device=DEV)
        dsts = [flat[i * per : (i + 1) * per] for i in range(ntensors)]
        us = time_fn(lambda: ops.chunked_copy(srcs, dsts), reps=20)
        gbs = 2 * total_mb / 1024 / (us / 1e6)  # read+write
        print(f'  {total_mb:5d} MB in {ntensors:4d} tensors: {us:9.1f} us  -> {gbs:7.0f} GB/s (r+w)')


def bench_reduce():
    print('== metric_reduce_into (full reduction into scalar acc) ==')
    for n in [1, 1024, 1 << 20, 1 << 26]:
        v = torch.randn(n, device=DEV)
        acc = torch.zeros(1, dtype=torch.float64, device=DEV)
        cnt = torch.zeros(1, dtype=torch.int64, device=DEV)
        us = time_fn(lambda: ops.metric_reduce_into(v, acc, cnt, ops.OP_SUM))
        gbs = n * 4 / (us / 1e6) / 1e9
        print(f'  n={n:>10}: {us:8.1f} us  ({gbs:7.1f} GB/s)')


def bench_adam():
    print('== fused_adam ==')
    for n in [1 << 16, 1 << 22, 1 << 26]:
        p = torch.randn(n, device=DEV)
        g = torch.randn(n, device=DEV)
        m = torch.zeros(n, device=DEV)
        v = torch.zeros(n, device=DEV)
        st = torch.zeros(1, dtype=torch.int32, device=DEV)
        us = time_fn(lambda: ops.fused_adam(p, g, m, v, st, 1e-3, 0.9, 0.999, 1e-8, 0.0))
        gbs = n * 4 * 7 / (us / 1e6) / 1e9  # 4 reads + 3 writes
        print(f'  n={n:>10}: {us:8.1f} us  ({gbs:7.1f} GB/s of 7x traffic)')


def bench_clip():
    print('== clip_grad_norm_ (norm + scale) ==')
    for n in [1 << 22, 1 << 26]:
        g = torch.randn(n, device=DEV)
        us = time_fn(lambda: ops.clip_grad_norm_(g, 1e9))
        gbs = n * 4 * 3 / (us / 1e6) / 1e9  # 2 reads + 1 write
        print(f'  n={n:>10}: {us:8.1f} us  ({gbs:7.1f} GB/s of 3x traffic)')


if __name__ == '__main__':
    bench_copy()
    bench_reduce()
    bench_adam()
    bench_clip()
