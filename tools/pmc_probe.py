"""Tiny workload for rocprofv3 --pmc counter collection: runs each fused
smallcnn kernel and the metric reducer a handful of times (small DB)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from dmlcloud_amd import _C, ops

DEV = 'cuda:0'


def main():
    n = 4096
    torch.manual_seed(0)
    for cin, cout, hw in [(1, 16, 28), (16, 16, 14)]:
        x = torch.randn(n, cin, hw, hw, device=DEV)
        w = torch.randn(cout, cin, 3, 3, device=DEV) * 0.1
        b = torch.randn(cout, device=DEV) * 0.1
        pooled = torch.empty(n, cout, hw // 2, hw // 2, device=DEV)
        argmax = torch.empty_like(pooled, dtype=torch.uint8)
        dpooled = torch.randn_like(pooled)
        din = torch.empty_like(x)
        dw = torch.zeros_like(w)
        db = torch.zeros_like(b)
        for _ in range(3):
            _C.conv3x3_relu_pool_fwd(x, w, b, pooled, argmax)
            _C.conv3x3_relu_pool_bwd_data(dpooled, argmax, pooled, w, din)
            _C.conv3x3_relu_pool_bwd_weight(dpooled, argmax, pooled, x, dw, db)

    v = torch.randn(1 << 22, device=DEV)
    acc = torch.zeros(1, dtype=torch.float64, device=DEV)
    cnt = torch.zeros(1, dtype=torch.int64, device=DEV)
    for _ in range(3):
        ops.metric_reduce_into(v, acc, cnt, ops.OP_SUM)

    p = torch.randn(1 << 22, device=DEV)
    g = torch.randn(1 << 22, device=DEV)
    m = torch.zeros(1 << 22, device=DEV)
    vv = torch.zeros(1 << 22, device=DEV)
    st = torch.zeros(1, dtype=torch.int32, device=DEV)
    for _ in range(3):
        ops.fused_adam(p, g, m, vv, st, 1e-3, 0.9, 0.999, 1e-8, 0.0)

    torch.cuda.synchronize()
    print('pmc probe done')


if __name__ == '__main__':
    main()
