"""Micro-benchmark of the fused smallcnn kernels per layer shape.

Run on a GPU box:  python tools/bench_smallcnn.py [batch ...]
Prints per-kernel times (cuda events, 100 reps) for the two MNIST-CNN
layer shapes, plus the torch/MIOpen equivalents for comparison.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from dmlcloud_amd import _C

DEV = 'cuda:0'


def time_fn(fn, reps=100, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(reps):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / reps * 1e3  # us


def bench_shape(n, cin, cout, hw):
    torch.manual_seed(0)
    x = torch.randn(n, cin, hw, hw, device=DEV)
    w = torch.randn(cout, cin, 3, 3, device=DEV) * 0.1
    b = torch.randn(cout, device=DEV) * 0.1
    pooled = torch.empty(n, cout, hw // 2, hw // 2, device=DEV)
    argmax = torch.empty(n, cout, hw // 2, hw // 2, device=DEV, dtype=torch.uint8)
    _C.conv3x3_relu_pool_fwd(x, w, b, pooled, argmax)
    dpooled = torch.randn_like(pooled)
    din = torch.empty_like(x)
    dw = torch.zeros_like(w)
    db = torch.zeros_like(b)

    t_fwd = time_fn(lambda: _C.conv3x3_relu_pool_fwd(x, w, b, pooled, argmax))
    t_bwdd = time_fn(lambda: _C.conv3x3_relu_pool_bwd_data(dpooled, argmax, pooled, w, din))
    t_bwdw = time_fn(lambda: _C.conv3x3_relu_pool_bwd_weight(dpooled, argmax, pooled, x, dw, db))

    # torch eager equivalents
    def torch_fwd():
        F.max_pool2d(F.relu(F.conv2d(x, w, b, padding=1)), 2)

    x2 = x.detach().requires_grad_(True)
    w2 = w.detach().requires_grad_(True)
    b2 = b.detach().requires_grad_(True)

    def torch_fwdbwd():
        out = F.max_pool2d(F.relu(F.conv2d(x2, w2, b2, padding=1)), 2)
        out.backward(dpooled)
        x2.grad = None
        w2.grad = None
        b2.grad = None

    t_tfwd = time_fn(torch_fwd, reps=50)
    t_tfb = time_fn(torch_fwdbwd, reps=50)

    print(
        f'N={n:6d} cin={cin:2d} cout={cout:2d} hw={hw:2d} | '
        f'fwd {t_fwd:8.1f}us bwd_data {t_bwdd:8.1f}us bwd_w {t_bwdw:8.1f}us '
        f'(sum {t_fwd + t_bwdd + t_bwdw:8.1f}us) | torch fwd {t_tfwd:8.1f}us fwd+bwd {t_tfb:8.1f}us'
    )


if __name__ == '__main__':
    batches = [int(a) for a in sys.argv[1:]] or [1024, 4096, 16384]
    for n in batches:
        bench_shape(n, 1, 16, 28)  # conv1 shape
        bench_shape(n, 16, 16, 14)  # conv2 shape
