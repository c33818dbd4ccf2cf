"""CPU-side semantics of the op layer (the same suite runs against the
gfx950 kernels in test_gpu.py)."""

import sys

import pytest
import torch

from dmlcloud_amd import ops


class TestFusedAdamReference:
    def test_matches_torch_adam(self):
        """Flat Adam reference implementation == torch.optim.Adam."""
        torch.manual_seed(0)
        n = 1003
        param_a = torch.randn(n)
        param_b = param_a.clone().requires_grad_(True)

        exp_avg = torch.zeros(n)
        exp_avg_sq = torch.zeros(n)
        step_t = torch.zeros(1, dtype=torch.int32)

        opt = torch.optim.Adam([param_b], lr=1e-2, betas=(0.9, 0.999), eps=1e-8)

        for it in range(5):
            torch.manual_seed(100 + it)
            grad = torch.randn(n)
            ops.fused_adam(param_a, grad, exp_avg, exp_avg_sq, step_t, 1e-2, 0.9, 0.999, 1e-8, 0.0)
            param_b.grad = grad.clone()
            opt.step()
            torch.testing.assert_close(param_a, param_b.detach(), rtol=1e-5, atol=1e-6)

    def test_weight_decay(self):
        torch.manual_seed(0)
        n = 64
        param_a = torch.randn(n)
        param_b = param_a.clone()
        exp_avg, exp_avg_sq = torch.zeros(n), torch.zeros(n)
        step_t = torch.zeros(1, dtype=torch.int32)
        grad = torch.randn(n)

        ops.fused_adam(param_a, grad, exp_avg, exp_avg_sq, step_t, 1e-2, 0.9, 0.999, 1e-8, 0.1)

        # manual L2-regularized Adam step
        g = grad + 0.1 * param_b
        m = 0.1 * g
        v = 0.001 * g * g
        denom = (v / (1 - 0.999)).sqrt() + 1e-8
        expected = param_b - 1e-2 * (m / (1 - 0.9)) / denom
        torch.testing.assert_close(param_a, expected, rtol=1e-5, atol=1e-6)

    def test_grad_scale(self):
        n = 32
        param_a = torch.ones(n)
        param_b = torch.ones(n)
        ea, eas = torch.zeros(n), torch.zeros(n)
        eb, ebs = torch.zeros(n), torch.zeros(n)
        sa = torch.zeros(1, dtype=torch.int32)
        sb = torch.zeros(1, dtype=torch.int32)
        grad = torch.full((n,), 2.0)
        ops.fused_adam(param_a, grad, ea, eas, sa, 1e-2, 0.9, 0.999, 1e-8, 0.0, grad_scale=0.5)
        ops.fused_adam(param_b, grad * 0.5, eb, ebs, sb, 1e-2, 0.9, 0.999, 1e-8, 0.0, grad_scale=1.0)
        torch.testing.assert_close(param_a, param_b)


class TestFusedAdamBf16Reference:
    def test_master_tracks_fp32_adam(self):
        """bf16 Adam's fp32 master must follow a plain fp32 Adam driven by
        the (bf16-rounded) gradients."""
        torch.manual_seed(0)
        n = 511
        master = torch.randn(n)
        param = master.to(torch.bfloat16)
        m1, v1 = torch.zeros(n), torch.zeros(n)
        st1 = torch.zeros(1, dtype=torch.int32)

        ref_p = master.clone()
        m2, v2 = torch.zeros(n), torch.zeros(n)
        st2 = torch.zeros(1, dtype=torch.int32)

        for it in range(3):
            torch.manual_seed(it)
            grad32 = torch.randn(n)
            grad16 = grad32.to(torch.bfloat16)
            ops.fused_adam_bf16(param, grad16, master, m1, v1, st1, 1e-2, 0.9, 0.999, 1e-8, 0.0)
            ops.fused_adam(ref_p, grad16.to(torch.float32), m2, v2, st2, 1e-2, 0.9, 0.999, 1e-8, 0.0)
        torch.testing.assert_close(master, ref_p, rtol=1e-6, atol=1e-7)
        torch.testing.assert_close(param, master.to(torch.bfloat16))


class TestFlatReplicaBf16:
    def test_bf16_views_and_master(self):
        from dmlcloud_amd.parallel import FlatAdam, FlatReplica

        torch.manual_seed(0)
        model = torch.nn.Linear(8, 4)
        w0 = model.weight.detach().clone()
        replica = FlatReplica(model, broadcast=False, dtype=torch.bfloat16)
        assert replica.flat_param.dtype == torch.bfloat16
        assert replica.flat_master.dtype == torch.float32
        assert model.weight.dtype == torch.bfloat16
        torch.testing.assert_close(model.weight.detach().float(), w0, rtol=1e-2, atol=1e-2)

        opt = FlatAdam(replica, lr=1e-2)
        x = torch.randn(4, 8, dtype=torch.bfloat16)
        replica.zero_grad()
        replica(x).float().pow(2).mean().backward()
        assert model.weight.grad.dtype == torch.bfloat16
        opt.step()
        # params and master moved together
        torch.testing.assert_close(replica.flat_param.float(), replica.flat_master, rtol=1e-2, atol=1e-2)


class TestFusedSgdReference:
    def test_matches_torch_sgd_momentum(self):
        torch.manual_seed(0)
        n = 517
        param_a = torch.randn(n)
        param_b = param_a.clone().requires_grad_(True)
        momentum_buf = torch.zeros(n)
        opt = torch.optim.SGD([param_b], lr=0.1, momentum=0.9)

        for it in range(4):
            torch.manual_seed(it)
            grad = torch.randn(n)
            ops.fused_sgd(param_a, grad, momentum_buf, 0.1, 0.9, 0.0)
            param_b.grad = grad.clone()
            opt.step()
            torch.testing.assert_close(param_a, param_b.detach(), rtol=1e-5, atol=1e-6)

    def test_plain_sgd(self):
        param = torch.ones(8)
        grad = torch.full((8,), 0.5)
        ops.fused_sgd(param, grad, None, 0.1, 0.0, 0.0)
        torch.testing.assert_close(param, torch.full((8,), 1.0 - 0.05))


class TestClip:
    def test_l2_norm(self):
        x = torch.tensor([3.0, 4.0])
        assert ops.l2_norm(x).item() == pytest.approx(5.0)

    def test_clip_applies(self):
        x = torch.tensor([3.0, 4.0])
        ops.clip_grad_norm_(x, 1.0)
        assert x.norm().item() == pytest.approx(1.0, rel=1e-4)

    def test_clip_noop_below_threshold(self):
        x = torch.tensor([0.3, 0.4])
        ops.clip_grad_norm_(x, 1.0)
        torch.testing.assert_close(x, torch.tensor([0.3, 0.4]))

    def test_matches_torch_clip(self):
        torch.manual_seed(0)
        g1 = torch.randn(1000)
        g2 = g1.clone().requires_grad_(False)
        ops.clip_grad_norm_(g1, 0.7)
        p = torch.nn.Parameter(torch.zeros(1000))
        p.grad = g2
        torch.nn.utils.clip_grad_norm_([p], 0.7)
        torch.testing.assert_close(g1, p.grad, rtol=1e-4, atol=1e-6)

    def test_norm_scale_emulates_world_averaging(self):
        """Clipping a W-summed gradient with norm_scale=1/W == clipping
        the averaged gradient then re-summing (the flat-path contract)."""
        torch.manual_seed(1)
        world = 4
        g_avg = torch.randn(1000)
        g_sum = g_avg * world

        norm = ops.clip_grad_norm_(g_sum, 0.5, norm_scale=1.0 / world)
        assert norm[0].item() == pytest.approx(g_avg.norm().item(), rel=1e-4)

        p = torch.nn.Parameter(torch.zeros(1000))
        p.grad = g_avg.clone()
        torch.nn.utils.clip_grad_norm_([p], 0.5)
        torch.testing.assert_close(g_sum / world, p.grad, rtol=1e-4, atol=1e-6)

    def test_l2_norm_scale(self):
        x = torch.tensor([3.0, 4.0])
        assert ops.l2_norm(x, norm_scale=0.5).item() == pytest.approx(2.5)


class TestChunkedCopy:
    def test_cpu_pack(self):
        srcs = [torch.randn(10), torch.randn(3, 4)]
        flat = torch.zeros(22)
        dsts = [flat[:10], flat[10:22]]
        ops.chunked_copy(srcs, dsts)
        torch.testing.assert_close(flat[:10], srcs[0])
        torch.testing.assert_close(flat[10:].reshape(3, 4), srcs[1])


if __name__ == '__main__':
    sys.exit(pytest.main([__file__]))
