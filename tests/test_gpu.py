"""gfx950 kernel numerics and GPU end-to-end paths.

Every test compares the HIP kernels against the pure-torch reference
implementations (ops/_reference.py) or against plain fp32 torch ops.
Run on an MI355X box: python -m pytest tests -m gpu
"""

import sys

import pytest
import torch

from dmlcloud_amd import ops
from dmlcloud_amd.ops import OP_MAX, OP_MIN, OP_SUM
from dmlcloud_amd.ops import _reference as ref

pytestmark = pytest.mark.gpu

DEV = 'cuda:0'


def _assert_native_loaded():
    assert ops.is_available(), 'native extension must be present on a GPU box'


class TestReduceKernels:
    @pytest.mark.parametrize('op', [OP_SUM, OP_MIN, OP_MAX])
    @pytest.mark.parametrize('n', [1, 7, 256, 16384, 16385, 1 << 20])
    @pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16, torch.float16])
    def test_reduce_into_float(self, op, n, dtype):
        _assert_native_loaded()
        torch.manual_seed(n)
        value = torch.randn(n, dtype=torch.float32)
        value_dev = value.to(DEV, dtype)

        acc_ref = torch.zeros(1, dtype=torch.float64)
        cnt_ref = torch.zeros(1, dtype=torch.int64)
        ref.reduce_into_acc(value.to(dtype), acc_ref, cnt_ref, op)

        acc = torch.zeros(1, dtype=torch.float64, device=DEV)
        if op == OP_MIN:
            acc.fill_(float('inf'))
            acc_ref2 = torch.full((1,), float('inf'), dtype=torch.float64)
            ref.reduce_into_acc(value.to(dtype), acc_ref2, torch.zeros(1, dtype=torch.int64), op)
            acc_ref = acc_ref2
        elif op == OP_MAX:
            acc.fill_(float('-inf'))
            acc_ref2 = torch.full((1,), float('-inf'), dtype=torch.float64)
            ref.reduce_into_acc(value.to(dtype), acc_ref2, torch.zeros(1, dtype=torch.int64), op)
            acc_ref = acc_ref2
        cnt = torch.zeros(1, dtype=torch.int64, device=DEV)
        ops.metric_reduce_into(value_dev, acc, cnt, op)

        assert cnt.item() == 1
        torch.testing.assert_close(acc.cpu(), acc_ref, rtol=1e-10, atol=1e-8)

    @pytest.mark.parametrize('op', [OP_SUM, OP_MIN, OP_MAX])
    def test_reduce_into_int64(self, op):
        _assert_native_loaded()
        value = torch.randint(-1000, 1000, (4096,), dtype=torch.int64)
        init = {OP_SUM: 0, OP_MIN: torch.iinfo(torch.int64).max, OP_MAX: torch.iinfo(torch.int64).min}[op]
        acc = torch.full((1,), init, dtype=torch.int64, device=DEV)
        cnt = torch.zeros(1, dtype=torch.int64, device=DEV)
        ops.metric_reduce_into(value.to(DEV), acc, cnt, op)
        expected = {OP_SUM: value.sum(), OP_MIN: value.min(), OP_MAX: value.max()}[op]
        assert acc.item() == expected.item()

    def test_deterministic(self):
        """Same input -> bitwise-identical accumulator (no atomics)."""
        _assert_native_loaded()
        value = torch.randn(1 << 20, device=DEV)
        results = []
        for _ in range(3):
            acc = torch.zeros(1, dtype=torch.float64, device=DEV)
            cnt = torch.zeros(1, dtype=torch.int64, device=DEV)
            ops.metric_reduce_into(value, acc, cnt, OP_SUM)
            results.append(acc.item())
        assert results[0] == results[1] == results[2]

    @pytest.mark.parametrize('op', [OP_SUM, OP_MIN, OP_MAX])
    def test_elementwise_accumulate(self, op):
        _assert_native_loaded()
        torch.manual_seed(0)
        values = [torch.randn(4, 5) for _ in range(3)]

        acc_ref = torch.zeros(4, 5, dtype=torch.float64)
        cnt_ref = torch.zeros(1, dtype=torch.int64)
        acc_dev = torch.zeros(4, 5, dtype=torch.float64, device=DEV)
        cnt_dev = torch.zeros(1, dtype=torch.int64, device=DEV)
        if op != OP_SUM:
            fill = float('inf') if op == OP_MIN else float('-inf')
            acc_ref.fill_(fill)
            acc_dev.fill_(fill)
        for v in values:
            ref.accumulate_elementwise(v, acc_ref, cnt_ref, op)
            ops.metric_accumulate_elementwise(v.to(DEV), acc_dev, cnt_dev, op)
        assert cnt_dev.item() == 3
        torch.testing.assert_close(acc_dev.cpu(), acc_ref)

    @pytest.mark.parametrize('op', [OP_SUM, OP_MIN, OP_MAX])
    @pytest.mark.parametrize('dims', [[0], [1], [0, 2], [1, 2], [0, 1, 2]])
    def test_finalize_dims(self, op, dims):
        _assert_native_loaded()
        torch.manual_seed(1)
        acc = torch.randn(3, 4, 5, dtype=torch.float64)
        out_ref = ref.finalize_dims(acc, dims, op)
        out_dev = ops.metric_finalize_dims(acc.to(DEV), dims, op)
        torch.testing.assert_close(out_dev.cpu(), out_ref)


class TestMetricsEndToEndGPU:
    def test_reducer_parity_with_cpu(self):
        """Same appended values -> same result on GPU accumulators as on
        the CPU torch path."""
        from dmlcloud_amd.metrics import MetricReducer, Reduction

        torch.manual_seed(0)
        values = [torch.randn(128) for _ in range(10)]
        for reduction in [Reduction.MEAN, Reduction.SUM, Reduction.MIN, Reduction.MAX]:
            r_cpu = MetricReducer(reduction)
            r_gpu = MetricReducer(reduction)
            for v in values:
                r_cpu.append(v)
                r_gpu.append(v.to(DEV))
            a = r_cpu.reduce_locally()
            b = r_gpu.reduce_locally().cpu()
            torch.testing.assert_close(a, b, rtol=1e-9, atol=1e-9)

    def test_tracker_epoch_gpu(self, torch_distributed_cuda):
        from dmlcloud_amd.metrics import MetricTracker, Reduction

        t = MetricTracker()
        t.register_metric('loss', Reduction.MEAN)
        for i in range(5):
            t.track('loss', torch.full((1,), float(i), device=DEV))
        t.next_epoch()
        assert t['loss'][0].item() == pytest.approx(2.0)


class TestOptimKernels:
    def test_adam_matches_reference(self):
        _assert_native_loaded()
        torch.manual_seed(0)
        n = (1 << 16) + 3  # odd tail exercises the scalar path
        param_ref = torch.randn(n)
        param_dev = param_ref.to(DEV)
        ea_ref, eas_ref = torch.zeros(n), torch.zeros(n)
        ea_dev, eas_dev = torch.zeros(n, device=DEV), torch.zeros(n, device=DEV)
        st_ref = torch.zeros(1, dtype=torch.int32)
        st_dev = torch.zeros(1, dtype=torch.int32, device=DEV)
        for it in range(3):
            torch.manual_seed(it)
            grad = torch.randn(n)
            ref.fused_adam_step(param_ref, grad, ea_ref, eas_ref, st_ref, 1e-2, 0.9, 0.999, 1e-8, 0.01, 0.5)
            ops.fused_adam(param_dev, grad.to(DEV), ea_dev, eas_dev, st_dev, 1e-2, 0.9, 0.999, 1e-8, 0.01, 0.5)
        torch.testing.assert_close(param_dev.cpu(), param_ref, rtol=1e-4, atol=1e-6)

    def test_sgd_matches_reference(self):
        _assert_native_loaded()
        torch.manual_seed(0)
        n = 12345
        param_ref = torch.randn(n)
        param_dev = param_ref.to(DEV)
        mom_ref = torch.zeros(n)
        mom_dev = torch.zeros(n, device=DEV)
        for it in range(3):
            torch.manual_seed(10 + it)
            grad = torch.randn(n)
            ref.fused_sgd_step(param_ref, grad, mom_ref, 0.1, 0.9, 0.001, 1.0)
            ops.fused_sgd(param_dev, grad.to(DEV), mom_dev, 0.1, 0.9, 0.001, 1.0)
        torch.testing.assert_close(param_dev.cpu(), param_ref, rtol=1e-5, atol=1e-6)

    def test_adam_bf16_matches_reference(self):
        _assert_native_loaded()
        torch.manual_seed(0)
        n = (1 << 16) + 5
        master_ref = torch.randn(n)
        param_ref = master_ref.to(torch.bfloat16)
        m_ref, v_ref = torch.zeros(n), torch.zeros(n)
        st_ref = torch.zeros(1, dtype=torch.int32)

        master_dev = master_ref.to(DEV)
        param_dev = param_ref.to(DEV)
        m_dev, v_dev = torch.zeros(n, device=DEV), torch.zeros(n, device=DEV)
        st_dev = torch.zeros(1, dtype=torch.int32, device=DEV)

        for it in range(3):
            torch.manual_seed(it)
            grad = torch.randn(n).to(torch.bfloat16)
            ref.fused_adam_bf16_step(param_ref, grad, master_ref, m_ref, v_ref, st_ref, 1e-2, 0.9, 0.999, 1e-8, 0.01, 0.5)
            ops.fused_adam_bf16(param_dev, grad.to(DEV), master_dev, m_dev, v_dev, st_dev, 1e-2, 0.9, 0.999, 1e-8, 0.01, 0.5)
        torch.testing.assert_close(master_dev.cpu(), master_ref, rtol=1e-4, atol=1e-6)
        torch.testing.assert_close(param_dev.cpu(), param_ref)

    def test_clip_bf16(self):
        _assert_native_loaded()
        g = torch.randn(1 << 20, device=DEV).to(torch.bfloat16)
        g_ref = g.float().cpu()
        norm = ops.clip_grad_norm_(g, 0.5)
        expected_norm = g_ref.norm()
        assert norm.item() == pytest.approx(expected_norm.item(), rel=1e-2)
        assert g.float().norm().item() == pytest.approx(0.5, rel=1e-2)

    def test_clip_matches_torch(self):
        _assert_native_loaded()
        torch.manual_seed(0)
        g = torch.randn(1 << 20, device=DEV)
        g_ref = g.cpu().clone()
        norm = ops.clip_grad_norm_(g, 0.5)
        p = torch.nn.Parameter(torch.zeros_like(g_ref))
        p.grad = g_ref
        norm_ref = torch.nn.utils.clip_grad_norm_([p], 0.5)
        # the gfx950 kernel accumulates in fp64; torch's fp32 norm carries
        # ~1e-5 relative accumulation error at 1M elements
        assert norm.item() == pytest.approx(norm_ref.item(), rel=1e-4)
        torch.testing.assert_close(g.cpu(), p.grad, rtol=1e-4, atol=1e-7)

    def test_clip_norm_scale(self):
        """norm_scale emulates the 1/world averaging: clipping a W-summed
        gradient with norm_scale=1/W == clipping the averaged gradient."""
        _assert_native_loaded()
        torch.manual_seed(3)
        world = 4
        g_avg = torch.randn(100_000, device=DEV)
        g_sum = g_avg * world

        norm = ops.clip_grad_norm_(g_sum, 0.5, norm_scale=1.0 / world)
        # reported norm is the averaged norm
        assert norm.item() == pytest.approx(g_avg.norm().item(), rel=1e-4)

        p = torch.nn.Parameter(torch.zeros_like(g_avg))
        p.grad = g_avg.clone()
        torch.nn.utils.clip_grad_norm_([p], 0.5)
        # after the later 1/W averaging, the clipped sum equals torch's clipped average
        torch.testing.assert_close(g_sum / world, p.grad, rtol=1e-4, atol=1e-7)


class TestCopyKernel:
    def test_pack_unpack_roundtrip(self):
        _assert_native_loaded()
        torch.manual_seed(0)
        tensors = [torch.randn(n, device=DEV) for n in (1024, 17, 1 << 20)]
        total = sum(t.numel() for t in tensors)
        flat = torch.zeros(total, device=DEV)
        offs = []
        off = 0
        for t in tensors:
            offs.append(off)
            off += t.numel()
        ops.chunked_copy(tensors, [flat[o : o + t.numel()] for o, t in zip(offs, tensors)])
        back = [torch.zeros_like(t) for t in tensors]
        ops.chunked_copy([flat[o : o + t.numel()] for o, t in zip(offs, tensors)], back)
        torch.cuda.synchronize()
        for t, b in zip(tensors, back):
            torch.testing.assert_close(t, b)

    def test_interleave_gpu_matches_cpu(self):
        from dmlcloud_amd.data import interleave_batches

        torch.manual_seed(0)
        batches = [torch.randn(64, 7) for _ in range(4)]
        cpu_out = [b.clone() for b in interleave_batches(iter(batches), 4)]
        gpu_out = [b.cpu().clone() for b in interleave_batches(iter([b.to(DEV) for b in batches]), 4)]
        for a, b in zip(cpu_out, gpu_out):
            torch.testing.assert_close(a, b)

    def test_dmlt_gpu_roundtrip(self, tmp_path):
        from dmlcloud_amd.checkpoint import load_tensor_state, save_tensor_state

        state = {'w': torch.randn(1000, 10, device=DEV), 'b': torch.randn(10, device=DEV), 'meta': 3}
        save_tensor_state(state, tmp_path / 's.dmlt')
        loaded = load_tensor_state(tmp_path / 's.dmlt', device=DEV)
        assert loaded['w'].device.type == 'cuda'
        torch.testing.assert_close(loaded['w'], state['w'])
        torch.testing.assert_close(loaded['b'], state['b'])


class TestFlatReplicaGPU:
    def test_train_convergence(self, torch_distributed_cuda):
        from dmlcloud_amd.parallel import FlatAdam, FlatReplica

        torch.manual_seed(0)
        model = torch.nn.Sequential(torch.nn.Linear(16, 64), torch.nn.ReLU(), torch.nn.Linear(64, 1)).to(DEV)
        replica = FlatReplica(model)
        opt = FlatAdam(replica, lr=1e-2)
        x = torch.randn(256, 16, device=DEV)
        y = x.sum(dim=1, keepdim=True)
        first_loss = None
        for _ in range(50):
            replica.zero_grad()
            loss = torch.nn.functional.mse_loss(replica(x), y)
            loss.backward()
            replica.grad_sync()
            opt.step()
            if first_loss is None:
                first_loss = loss.item()
        assert loss.item() < first_loss * 0.1

    def test_graphed_step(self, torch_distributed_cuda):
        from dmlcloud_amd.parallel import FlatReplica, FlatSGD, GraphedStep

        torch.manual_seed(0)
        model = torch.nn.Linear(32, 32).to(DEV)
        replica = FlatReplica(model)
        opt = FlatSGD(replica, lr=1e-3)
        x = torch.randn(64, 32, device=DEV)

        def step():
            replica.zero_grad()
            loss = replica(x).pow(2).mean()
            loss.backward()
            replica.grad_sync()
            opt.step()

        gs = GraphedStep(step, warmup=3)
        gs.initialize()
        assert gs.captured, 'hipGraph capture must succeed for the flat path'
        p0 = replica.flat_param.clone()
        for _ in range(5):
            gs()
        torch.cuda.synchronize()
        assert not torch.equal(p0, replica.flat_param)


class TestFusedCnn:
    """Fused conv3x3+relu+pool kernels vs the plain torch fp32 oracle."""

    @pytest.mark.parametrize('cin,cout,hw,n', [(1, 16, 28, 8), (16, 16, 14, 8), (3, 32, 32, 4), (5, 7, 12, 3)])
    def test_forward_matches_torch(self, cin, cout, hw, n):
        import torch.nn.functional as F

        from dmlcloud_amd.ops.fused_cnn import ConvReluPool2d

        torch.manual_seed(0)
        layer = ConvReluPool2d(cin, cout).to(DEV)
        x = torch.randn(n, cin, hw, hw, device=DEV)
        out = layer(x)
        ref_out = F.max_pool2d(F.relu(F.conv2d(x, layer.weight, layer.bias, padding=1)), 2)
        torch.testing.assert_close(out, ref_out, rtol=1e-5, atol=1e-5)

    @pytest.mark.parametrize('cin,cout,hw,n', [(1, 16, 28, 4), (16, 16, 14, 4), (4, 8, 8, 2)])
    def test_backward_matches_torch(self, cin, cout, hw, n):
        import torch.nn.functional as F

        from dmlcloud_amd.ops.fused_cnn import ConvReluPool2d

        torch.manual_seed(1)
        layer = ConvReluPool2d(cin, cout).to(DEV)
        x = torch.randn(n, cin, hw, hw, device=DEV, requires_grad=True)
        out = layer(x)
        g = torch.randn_like(out)
        out.backward(g)

        w2 = layer.weight.detach().clone().requires_grad_(True)
        b2 = layer.bias.detach().clone().requires_grad_(True)
        x2 = x.detach().clone().requires_grad_(True)
        ref = F.max_pool2d(F.relu(F.conv2d(x2, w2, b2, padding=1)), 2)
        ref.backward(g)

        torch.testing.assert_close(layer.weight.grad, w2.grad, rtol=1e-4, atol=1e-4)
        torch.testing.assert_close(layer.bias.grad, b2.grad, rtol=1e-4, atol=1e-4)
        torch.testing.assert_close(x.grad, x2.grad, rtol=1e-4, atol=1e-4)

    def test_full_model_matches_eager(self):
        """FusedMnistCNN step == plain-torch mnist_cnn step (same weights)."""
        from dmlcloud_amd.models import mnist_cnn
        from dmlcloud_amd.ops.fused_cnn import FusedMnistCNN

        torch.manual_seed(0)
        fused = FusedMnistCNN().to(DEV)
        eager = mnist_cnn().to(DEV)
        with torch.no_grad():
            eager[0].weight.copy_(fused.layer1.weight)
            eager[0].bias.copy_(fused.layer1.bias)
            eager[3].weight.copy_(fused.layer2.weight)
            eager[3].bias.copy_(fused.layer2.bias)
            eager[7].weight.copy_(fused.fc.weight)
            eager[7].bias.copy_(fused.fc.bias)

        x = torch.randn(32, 1, 28, 28, device=DEV)
        y = torch.randint(0, 10, (32,), device=DEV)
        loss_f = torch.nn.functional.cross_entropy(fused(x), y)
        loss_e = torch.nn.functional.cross_entropy(eager(x), y)
        torch.testing.assert_close(loss_f, loss_e, rtol=1e-5, atol=1e-5)

        loss_f.backward()
        loss_e.backward()
        torch.testing.assert_close(fused.layer1.weight.grad, eager[0].weight.grad, rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(fused.layer2.weight.grad, eager[3].weight.grad, rtol=1e-4, atol=1e-5)


class TestFusedLayerNorm:
    @pytest.mark.parametrize('rows,d', [(64, 768), (1000, 768), (128, 1024), (37, 256)])
    def test_forward_matches_torch(self, rows, d):
        import torch.nn.functional as F

        torch.manual_seed(0)
        x = torch.randn(rows, d, device=DEV).to(torch.bfloat16)
        g = torch.randn(d, device=DEV).to(torch.bfloat16)
        b = torch.randn(d, device=DEV).to(torch.bfloat16)

        from dmlcloud_amd import _C

        y = torch.empty_like(x)
        mean = torch.empty(rows, dtype=torch.float32, device=DEV)
        rstd = torch.empty(rows, dtype=torch.float32, device=DEV)
        _C.layernorm_fwd(x, g, b, y, mean, rstd, 1e-5)
        ref_out = F.layer_norm(x.float(), (d,), g.float(), b.float(), 1e-5).to(torch.bfloat16)
        torch.testing.assert_close(y.float(), ref_out.float(), rtol=2e-2, atol=2e-2)

    def test_backward_matches_torch(self):
        import torch.nn.functional as F

        from dmlcloud_amd.ops.fused_ln import LayerNorm

        torch.manual_seed(1)
        rows, d = 512, 768
        ln = LayerNorm(d).to(DEV).to(torch.bfloat16)
        x = torch.randn(rows, d, device=DEV).to(torch.bfloat16).requires_grad_(True)
        out = ln(x)
        dy = torch.randn_like(out)
        out.backward(dy)

        x2 = x.detach().float().requires_grad_(True)
        g2 = ln.weight.detach().float().requires_grad_(True)
        b2 = ln.bias.detach().float().requires_grad_(True)
        ref = F.layer_norm(x2, (d,), g2, b2, 1e-5)
        ref.backward(dy.float())

        torch.testing.assert_close(x.grad.float(), x2.grad, rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(ln.weight.grad.float(), g2.grad, rtol=2e-2, atol=1.0)
        torch.testing.assert_close(ln.bias.grad.float(), b2.grad, rtol=2e-2, atol=1.0)


class TestFusedCE:
    @pytest.mark.parametrize('rows,v', [(128, 50257), (64, 512), (256, 1000)])
    def test_loss_matches_torch(self, rows, v):
        import torch.nn.functional as F

        from dmlcloud_amd.ops.fused_loss import cross_entropy

        torch.manual_seed(0)
        logits = (torch.randn(rows, v, device=DEV) * 3).to(torch.bfloat16)
        targets = torch.randint(0, v, (rows,), device=DEV)
        loss = cross_entropy(logits, targets)
        ref = F.cross_entropy(logits.float(), targets)
        assert loss.item() == pytest.approx(ref.item(), rel=2e-3)

    def test_grad_matches_torch(self):
        import torch.nn.functional as F

        from dmlcloud_amd.ops.fused_loss import cross_entropy

        torch.manual_seed(1)
        rows, v = 64, 50257
        logits = (torch.randn(rows, v, device=DEV) * 2).to(torch.bfloat16).requires_grad_(True)
        targets = torch.randint(0, v, (rows,), device=DEV)
        cross_entropy(logits, targets).backward()

        l2 = logits.detach().float().requires_grad_(True)
        F.cross_entropy(l2, targets).backward()
        torch.testing.assert_close(logits.grad.float(), l2.grad, rtol=5e-2, atol=1e-5)

    def test_ignore_index_matches_torch(self):
        """ignore_index rows contribute no loss/grad; mean divides by the
        valid count (padded-token safety)."""
        import torch.nn.functional as F

        from dmlcloud_amd.ops.fused_loss import cross_entropy

        torch.manual_seed(2)
        rows, v = 64, 1000
        logits = (torch.randn(rows, v, device=DEV) * 2).to(torch.bfloat16).requires_grad_(True)
        targets = torch.randint(0, v, (rows,), device=DEV)
        targets[::3] = -100  # every third row padded
        loss = cross_entropy(logits, targets)
        loss.backward()

        l2 = logits.detach().float().requires_grad_(True)
        ref = F.cross_entropy(l2, targets)
        ref.backward()
        assert loss.item() == pytest.approx(ref.item(), rel=2e-3)
        torch.testing.assert_close(logits.grad.float(), l2.grad, rtol=5e-2, atol=1e-5)
        # ignored rows have exactly zero gradient
        assert logits.grad[::3].abs().max().item() == 0.0

    def test_out_of_range_target_is_loud(self):
        """A target >= V (not ignore_index) produces NaN loss, not an OOB read."""
        from dmlcloud_amd.ops.fused_loss import cross_entropy

        logits = torch.randn(4, 100, device=DEV).to(torch.bfloat16)
        targets = torch.tensor([0, 1, 100, 2], device=DEV)  # 100 out of range
        loss = cross_entropy(logits, targets)
        assert torch.isnan(loss).item()

    def test_gpt2_tiny_step_matches_eager(self):
        """Full fused-GPT2 (LN + CE) loss matches a plain-torch computation."""
        from dmlcloud_amd.models import gpt2_tiny

        torch.manual_seed(0)
        model = gpt2_tiny().to(DEV)
        # run fp32 (fused paths disabled) vs bf16 (fused) on same weights
        idx = torch.randint(0, model.cfg.vocab_size, (2, 32), device=DEV)
        _, loss32 = model(idx, targets=idx)

        model16 = gpt2_tiny().to(DEV)
        model16.load_state_dict(model.state_dict())
        model16 = model16.to(torch.bfloat16)
        _, loss16 = model16(idx, targets=idx)
        assert loss16.item() == pytest.approx(loss32.item(), rel=5e-2)


class TestAttnFwdExperimental:
    """EXPERIMENTAL flash-attention forward (attention.hip) vs SDPA and
    the CPU tile blueprint."""

    @pytest.mark.parametrize('causal', [True, False])
    @pytest.mark.parametrize('b,h,n', [(1, 1, 64), (2, 3, 128), (2, 4, 1024)])
    def test_matches_sdpa(self, causal, b, h, n):
        import math

        from dmlcloud_amd import _C

        torch.manual_seed(0)
        d = 64
        q = (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16)
        k = (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16)
        v = (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16)
        o = torch.empty_like(q)
        lse = torch.empty(b, h, n, dtype=torch.float32, device=DEV)
        _C.attn_fwd(q, k, v, o, lse, 1.0 / math.sqrt(d), causal)

        ref = torch.nn.functional.scaled_dot_product_attention(
            q.float(), k.float(), v.float(), is_causal=causal
        )
        torch.testing.assert_close(o.float(), ref, rtol=3e-2, atol=3e-2)

    @pytest.mark.parametrize('causal', [True, False])
    @pytest.mark.parametrize('b,h,n', [(1, 1, 64), (2, 3, 128), (2, 4, 1024)])
    def test_backward_matches_sdpa(self, causal, b, h, n):
        from dmlcloud_amd.ops.fused_attn import sdpa

        torch.manual_seed(3)
        d = 64
        q = (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16).requires_grad_(True)
        k = (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16).requires_grad_(True)
        v = (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16).requires_grad_(True)
        do = (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16)

        out = sdpa(q, k, v, causal=causal)
        out.backward(do)

        q2 = q.detach().float().requires_grad_(True)
        k2 = k.detach().float().requires_grad_(True)
        v2 = v.detach().float().requires_grad_(True)
        ref = torch.nn.functional.scaled_dot_product_attention(q2, k2, v2, is_causal=causal)
        ref.backward(do.float())

        torch.testing.assert_close(out.float(), ref, rtol=3e-2, atol=3e-2)
        torch.testing.assert_close(q.grad.float(), q2.grad, rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(k.grad.float(), k2.grad, rtol=5e-2, atol=5e-2)
        torch.testing.assert_close(v.grad.float(), v2.grad, rtol=5e-2, atol=5e-2)

    def test_strided_views_no_copy(self):
        """Transpose views ([B,T,H,D] memory seen as [B,H,T,D]) run through
        the stride-aware kernels and match the contiguous path."""
        from dmlcloud_amd.ops.fused_attn import sdpa

        torch.manual_seed(5)
        b, h, n, d = 2, 3, 128, 64
        base_q = (torch.randn(b, n, h, d, device=DEV) * 0.5).to(torch.bfloat16)
        base_k = (torch.randn(b, n, h, d, device=DEV) * 0.5).to(torch.bfloat16)
        base_v = (torch.randn(b, n, h, d, device=DEV) * 0.5).to(torch.bfloat16)
        q = base_q.transpose(1, 2).requires_grad_(True)
        k = base_k.transpose(1, 2).requires_grad_(True)
        v = base_v.transpose(1, 2).requires_grad_(True)
        assert not q.is_contiguous()

        do = (torch.randn(b, h, n, d, device=DEV) * 0.5).to(torch.bfloat16)
        out = sdpa(q, k, v, causal=True)
        out.backward(do)

        q2 = q.detach().contiguous().requires_grad_(True)
        k2 = k.detach().contiguous().requires_grad_(True)
        v2 = v.detach().contiguous().requires_grad_(True)
        out2 = sdpa(q2, k2, v2, causal=True)
        out2.backward(do)

        torch.testing.assert_close(out, out2)
        torch.testing.assert_close(q.grad, q2.grad)
        torch.testing.assert_close(k.grad, k2.grad)
        torch.testing.assert_close(v.grad, v2.grad)

    def test_cross_attention_falls_back(self):
        """Different kv length (cross-attention geometry) must take the
        torch SDPA fallback, not feed the self-attention kernels
        (OOB read)."""
        from dmlcloud_amd.ops.fused_attn import sdpa

        torch.manual_seed(7)
        q = (torch.randn(2, 4, 128, 64, device=DEV) * 0.5).to(torch.bfloat16)
        k = (torch.randn(2, 4, 256, 64, device=DEV) * 0.5).to(torch.bfloat16)
        v = (torch.randn(2, 4, 256, 64, device=DEV) * 0.5).to(torch.bfloat16)
        out = sdpa(q, k, v, causal=False)
        ref = torch.nn.functional.scaled_dot_product_attention(q.float(), k.float(), v.float(), is_causal=False)
        torch.testing.assert_close(out.float(), ref, rtol=3e-2, atol=3e-2)

    def test_lse_matches_blueprint(self):
        import math

        from dmlcloud_amd import _C
        from dmlcloud_amd.ops._attention_ref import flash_attn_fwd_tiled

        torch.manual_seed(1)
        n, d = 128, 64
        q = (torch.randn(1, 1, n, d) * 0.5).to(torch.bfloat16)
        k = (torch.randn(1, 1, n, d) * 0.5).to(torch.bfloat16)
        v = (torch.randn(1, 1, n, d) * 0.5).to(torch.bfloat16)
        o = torch.empty_like(q).to(DEV)
        lse = torch.empty(1, 1, n, dtype=torch.float32, device=DEV)
        _C.attn_fwd(q.to(DEV), k.to(DEV), v.to(DEV), o, lse, 1.0 / math.sqrt(d), True)

        ref_o, ref_lse = flash_attn_fwd_tiled(q[0, 0].float(), k[0, 0].float(), v[0, 0].float(), causal=True)
        torch.testing.assert_close(lse.cpu()[0, 0], ref_lse, rtol=1e-2, atol=1e-2)
        torch.testing.assert_close(o.float().cpu()[0, 0], ref_o, rtol=3e-2, atol=3e-2)


class TestPipelineGPU:
    def test_smoke_gpu(self, torch_distributed_cuda):
        from dmlcloud_amd import TrainingPipeline, TrainValStage

        class DS(torch.utils.data.Dataset):
            def __len__(self):
                return 16

            def __getitem__(self, idx):
                g = torch.Generator().manual_seed(idx)
                return torch.randn(10, generator=g), idx % 10

        class Stage_(TrainValStage):
            def pre_stage(self):
                model = torch.nn.Linear(10, 10)
                self.pipeline.register_model('m', model)
                self.pipeline.register_optimizer('sgd', torch.optim.SGD(model.parameters(), lr=1e-2))
                self.pipeline.register_dataset('train', torch.utils.data.DataLoader(DS(), batch_size=4))
                self.pipeline.register_dataset('val', torch.utils.data.DataLoader(DS(), batch_size=4))
                self.loss = torch.nn.CrossEntropyLoss()

            def step(self, batch):
                x, y = batch
                x, y = x.to(self.device), y.to(self.device)
                return self.loss(self.pipeline.models['m'](x), y)

        pipeline = TrainingPipeline()
        pipeline.append_stage(Stage_(), max_epochs=2)
        pipeline.run()
        assert pipeline.device.type == 'cuda'
        assert pipeline.tracker['train/loss'][1] is not None


if __name__ == '__main__':
    sys.exit(pytest.main([__file__, '-m', 'gpu']))
