"""Metric system semantics (mirrors the reference's coverage, reference
test/test_metrics.py, plus accumulator-specific invariants)."""

import sys

import pytest
import torch

from dmlcloud_amd.metrics import MetricReducer, MetricTracker, Reduction, reduce_tensor


class TestReduceTensor:
    def test_full_reduction(self):
        t = torch.tensor([[1.0, 2.0], [3.0, 4.0]])
        assert reduce_tensor(t, Reduction.MEAN).item() == pytest.approx(2.5)
        assert reduce_tensor(t, Reduction.SUM).item() == pytest.approx(10.0)
        assert reduce_tensor(t, Reduction.MIN).item() == pytest.approx(1.0)
        assert reduce_tensor(t, Reduction.MAX).item() == pytest.approx(4.0)

    def test_partial_dims(self):
        t = torch.arange(24, dtype=torch.float32).reshape(2, 3, 4)
        assert reduce_tensor(t, Reduction.SUM, dim=[0]).shape == (3, 4)
        assert reduce_tensor(t, Reduction.MAX, dim=[1, 2]).shape == (2,)


class TestLocalReductions:
    def _fill(self, reduction, values):
        reducer = MetricReducer(reduction=reduction)
        for v in values:
            reducer.append(v)
        return reducer

    def test_mean(self):
        r = self._fill(Reduction.MEAN, [torch.tensor([1.0, 2.0]), torch.tensor([3.0, 4.0])])
        assert r.reduce_locally().item() == pytest.approx(2.5)

    def test_sum(self):
        r = self._fill(Reduction.SUM, [torch.tensor([1.0, 2.0]), torch.tensor([3.0, 4.0])])
        assert r.reduce_locally().item() == pytest.approx(10.0)

    def test_min(self):
        r = self._fill(Reduction.MIN, [torch.tensor([5.0, 2.0]), torch.tensor([3.0, 4.0])])
        assert r.reduce_locally().item() == pytest.approx(2.0)

    def test_max(self):
        r = self._fill(Reduction.MAX, [torch.tensor([5.0, 2.0]), torch.tensor([3.0, 4.0])])
        assert r.reduce_locally().item() == pytest.approx(5.0)

    def test_scalars_and_floats(self):
        r = self._fill(Reduction.MEAN, [1.0, 2.0, torch.tensor(3.0)])
        assert r.reduce_locally().item() == pytest.approx(2.0)

    def test_iadd(self):
        r = MetricReducer(Reduction.SUM)
        r += torch.tensor(1.0)
        r += torch.tensor(2.0)
        assert r.reduce_locally().item() == pytest.approx(3.0)
        assert len(r) == 2

    def test_empty_is_none(self):
        assert MetricReducer(Reduction.MEAN).reduce_locally() is None

    def test_integer_sum_exact(self):
        r = self._fill(Reduction.SUM, [torch.tensor(1), torch.tensor(1), torch.tensor(1)])
        out = r.reduce_locally()
        assert out.item() == 3

    def test_mean_int_raises(self):
        r = MetricReducer(Reduction.MEAN)
        with pytest.raises(RuntimeError):
            r.append(torch.tensor(1))

    def test_shape_mismatch_raises(self):
        r = MetricReducer(Reduction.MEAN)
        r.append(torch.zeros(3))
        with pytest.raises(ValueError):
            r.append(torch.zeros(4))

    def test_matches_stack_reference(self):
        """Differential test against the reference semantics
        (torch.stack + reduce over [0] + dim+1)."""
        torch.manual_seed(0)
        values = [torch.randn(4, 5) for _ in range(7)]
        for reduction, fn in [
            (Reduction.MEAN, lambda s: s.mean()),
            (Reduction.SUM, lambda s: s.sum()),
            (Reduction.MIN, lambda s: s.amin()),
            (Reduction.MAX, lambda s: s.amax()),
        ]:
            r = MetricReducer(reduction=reduction)
            r.extend(values)
            expected = fn(torch.stack(values))
            assert r.reduce_locally().item() == pytest.approx(expected.item(), rel=1e-6)


class TestPartialDimReductions:
    def test_dim0(self):
        # dim=[0] of the VALUE: stacked (N, 4, 5) reduced over stack dim + dim 1
        torch.manual_seed(1)
        values = [torch.randn(4, 5) for _ in range(3)]
        r = MetricReducer(reduction=Reduction.SUM, dim=[0])
        r.extend(values)
        expected = torch.stack(values).sum(dim=[0, 1])
        out = r.reduce_locally()
        assert out.shape == (5,)
        torch.testing.assert_close(out.to(expected.dtype), expected, rtol=1e-5, atol=1e-5)

    def test_dim1_mean(self):
        torch.manual_seed(2)
        values = [torch.randn(4, 5) for _ in range(3)]
        r = MetricReducer(reduction=Reduction.MEAN, dim=[1])
        r.extend(values)
        expected = torch.stack(values).mean(dim=[0, 2])
        out = r.reduce_locally()
        assert out.shape == (4,)
        torch.testing.assert_close(out.to(expected.dtype), expected, rtol=1e-5, atol=1e-5)

    def test_min_partial(self):
        torch.manual_seed(3)
        values = [torch.randn(4, 5) for _ in range(3)]
        r = MetricReducer(reduction=Reduction.MIN, dim=[0])
        r.extend(values)
        expected = torch.stack(values).amin(dim=[0, 1])
        torch.testing.assert_close(r.reduce_locally().to(expected.dtype), expected)

    def test_int_dim_normalized(self):
        r = MetricReducer(reduction=Reduction.SUM, dim=1)
        assert r.dim == [1]


class TestGlobalReduction:
    def test_world1(self, torch_distributed):
        r = MetricReducer(Reduction.MEAN)
        r.append(torch.tensor([2.0, 4.0]))
        assert r.reduce_globally().item() == pytest.approx(3.0)

    def test_world1_empty(self, torch_distributed):
        r = MetricReducer(Reduction.MEAN)
        assert r.reduce_globally() is None

    def test_not_globally(self, torch_distributed):
        r = MetricReducer(Reduction.SUM, globally=False)
        r.append(torch.tensor(5.0))
        assert r.reduce_globally().item() == pytest.approx(5.0)


class TestStateDict:
    def test_reducer_roundtrip(self):
        r = MetricReducer(Reduction.SUM, dim=[0])
        r.append(torch.ones(3, 2))
        r.append(torch.ones(3, 2))
        state = r.state_dict()
        r2 = MetricReducer()
        r2.load_state_dict(state)
        torch.testing.assert_close(r2.reduce_locally(), r.reduce_locally())

    def test_tracker_roundtrip(self, torch_distributed):
        t = MetricTracker()
        t.register_metric('a', Reduction.MEAN)
        t.register_metric('b')
        t.track('a', torch.tensor(1.0))
        t.track('b', 42)
        t.next_epoch()
        state = t.state_dict()

        t2 = MetricTracker()
        t2.load_state_dict(state)
        assert t2.epoch == 2
        assert t2['a'][0].item() == pytest.approx(1.0)
        assert t2['b'][0] == 42


class TestTracker:
    def test_register_and_track(self, torch_distributed):
        t = MetricTracker()
        t.register_metric('loss', Reduction.MEAN)
        assert 'loss' in t
        assert t.is_reduced_metric('loss')
        t.track('loss', torch.tensor(2.0))
        t.track('loss', torch.tensor(4.0))
        t.next_epoch()
        assert t['loss'][0].item() == pytest.approx(3.0)

    def test_missing_metric_raises(self):
        t = MetricTracker()
        with pytest.raises(ValueError):
            t.track('nope', 1)
        with pytest.raises(ValueError):
            t['nope']

    def test_double_register_raises(self):
        t = MetricTracker()
        t.register_metric('x')
        with pytest.raises(ValueError):
            t.register_metric('x')

    def test_double_track_nonreduced_raises(self, torch_distributed):
        t = MetricTracker()
        t.register_metric('x')
        t.track('x', 1)
        with pytest.raises(ValueError):
            t.track('x', 2)

    def test_strict_double_reduce_raises(self, torch_distributed):
        t = MetricTracker()
        t.register_metric('x')
        t.track('x', 1)
        t.reduce_all(strict=False)
        with pytest.raises(ValueError):
            t.reduce_all(strict=True)

    def test_late_registration_backfills(self, torch_distributed):
        t = MetricTracker()
        t.register_metric('a', Reduction.MEAN)
        t.track('a', 1.0)
        t.next_epoch()
        t.register_metric('late', Reduction.MEAN)
        t.track('late', 5.0)
        t.next_epoch()
        assert t['late'] == [None, torch.tensor(5.0)]

    def test_untracked_epoch_gives_none(self, torch_distributed):
        t = MetricTracker()
        t.register_metric('a', Reduction.MEAN)
        t.track('a', 1.0)
        t.next_epoch()
        t.next_epoch()  # nothing tracked
        history = t['a']
        assert history[1] is None

    def test_current_value_and_has_value(self, torch_distributed):
        t = MetricTracker()
        t.register_metric('a', Reduction.MEAN)
        assert t.current_value('a') is None
        t.track('a', 1.0)
        assert not t.has_value('a')
        t.reduce_all(strict=False)
        assert t.has_value('a')
        assert t.current_value('a').item() == pytest.approx(1.0)

    def test_dim_without_reduction_raises(self):
        t = MetricTracker()
        with pytest.raises(ValueError):
            t.register_metric('x', reduction=None, dim=[0])

    def test_epoch_filling_many(self, torch_distributed):
        t = MetricTracker()
        t.register_metric('m', Reduction.SUM)
        for epoch in range(5):
            for _ in range(3):
                t.track('m', torch.tensor(1.0))
            t.next_epoch()
        assert [h.item() for h in t['m']] == [3.0] * 5


if __name__ == '__main__':
    sys.exit(pytest.main([__file__]))
