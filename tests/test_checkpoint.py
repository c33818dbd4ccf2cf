"""Checkpoint directories and the .dmlt tensor-state format."""

import sys

import pytest
import torch

from dmlcloud_amd.checkpoint import (
    CheckpointDir,
    ModelCheckpointer,
    find_slurm_checkpoint,
    generate_checkpoint_path,
    generate_id,
    load_tensor_state,
    sanitize_filename,
    save_tensor_state,
)
from dmlcloud_amd.config import Config


class TestHelpers:
    def test_sanitize(self):
        assert sanitize_filename('a/b/c') == 'a_b_c'

    def test_generate_id(self):
        ids = {generate_id() for _ in range(50)}
        assert len(ids) == 50
        for i in ids:
            assert '-' not in i and '_' not in i

    def test_generate_path(self, tmp_path):
        p = generate_checkpoint_path(tmp_path, name='exp/1')
        assert p.parent == tmp_path
        assert p.name.startswith('exp_1-')


class TestCheckpointDir:
    def test_create_and_validity(self, tmp_path):
        d = CheckpointDir(tmp_path / 'run1')
        assert not d.is_valid
        d.create()
        assert d.is_valid
        assert d.indicator_file.exists()
        assert d.log_file.exists()

    def test_double_create_raises(self, tmp_path):
        d = CheckpointDir(tmp_path / 'run1')
        d.create()
        with pytest.raises(ValueError):
            d.create()

    def test_config_roundtrip(self, tmp_path):
        d = CheckpointDir(tmp_path / 'run1')
        d.create()
        cfg = Config.create({'lr': 0.1, 'nested': {'a': [1, 2]}})
        d.save_config(cfg)
        loaded = d.load_config()
        assert loaded.lr == 0.1
        assert loaded.nested.a == [1, 2]

    def test_slurm_discovery(self, tmp_path, monkeypatch):
        monkeypatch.setenv('SLURM_JOB_ID', '12345')
        d = CheckpointDir(tmp_path / 'run1')
        d.create()
        assert d.slurm_job_id == '12345'
        assert find_slurm_checkpoint(tmp_path) == tmp_path / 'run1'
        monkeypatch.setenv('SLURM_JOB_ID', '99999')
        assert find_slurm_checkpoint(tmp_path) is None


class TestDmltFormat:
    def test_roundtrip_nested(self, tmp_path):
        state = {
            'a': torch.randn(5, 3),
            'nested': {'b': torch.arange(7), 'c': [torch.ones(2), 'text', 3.14]},
            'scalar': 42,
            'flag': torch.tensor([True, False, True]),
            'half': torch.randn(4).to(torch.bfloat16),
        }
        path = tmp_path / 'state.dmlt'
        save_tensor_state(state, path)
        loaded = load_tensor_state(path)
        torch.testing.assert_close(loaded['a'], state['a'])
        torch.testing.assert_close(loaded['nested']['b'], state['nested']['b'])
        torch.testing.assert_close(loaded['nested']['c'][0], state['nested']['c'][0])
        assert loaded['nested']['c'][1] == 'text'
        assert loaded['scalar'] == 42
        assert loaded['flag'].tolist() == [True, False, True]
        torch.testing.assert_close(loaded['half'], state['half'])

    def test_model_state_roundtrip(self, tmp_path):
        model = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.BatchNorm1d(8), torch.nn.Linear(8, 2))
        path = tmp_path / 'model.dmlt'
        save_tensor_state(model.state_dict(), path)
        loaded = load_tensor_state(path)
        model2 = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.BatchNorm1d(8), torch.nn.Linear(8, 2))
        model2.load_state_dict(loaded)
        for p1, p2 in zip(model.parameters(), model2.parameters()):
            torch.testing.assert_close(p1, p2)

    def test_empty_state(self, tmp_path):
        path = tmp_path / 'empty.dmlt'
        save_tensor_state({'note': 'nothing'}, path)
        assert load_tensor_state(path) == {'note': 'nothing'}

    def test_bad_magic_raises(self, tmp_path):
        path = tmp_path / 'bad.dmlt'
        path.write_bytes(b'NOTDMLT0' + b'\x00' * 16)
        with pytest.raises(ValueError):
            load_tensor_state(path)


class TestModelCheckpointer:
    def _dir(self, tmp_path):
        d = CheckpointDir(tmp_path / 'run')
        d.create()
        return d

    def test_latest(self, tmp_path):
        d = self._dir(tmp_path)
        model = torch.nn.Linear(3, 3)
        ck = ModelCheckpointer(d, 'net', save_latest=True)
        ck.maybe_save(model, epoch=1)
        loaded = ck.load('latest')
        assert loaded['epoch'] == 1
        torch.testing.assert_close(loaded['state_dict']['weight'], model.weight)

    def test_interval(self, tmp_path):
        d = self._dir(tmp_path)
        model = torch.nn.Linear(3, 3)
        ck = ModelCheckpointer(d, 'net', save_latest=False, save_interval=2)
        for epoch in range(1, 5):
            ck.maybe_save(model, epoch=epoch)
        files = sorted(p.name for p in (d.models_dir / 'net').iterdir())
        assert files == ['epoch_0002.dmlt', 'epoch_0004.dmlt']

    def test_best(self, tmp_path):
        from dmlcloud_amd.metrics import MetricTracker, Reduction

        d = self._dir(tmp_path)
        model = torch.nn.Linear(3, 3)
        ck = ModelCheckpointer(d, 'net', save_latest=False, save_best=True, best_metric='val/loss')

        tracker = MetricTracker()
        tracker.register_metric('val/loss', Reduction.MEAN, globally=False)
        for epoch, loss in enumerate([3.0, 1.0, 2.0], start=1):
            tracker.track('val/loss', torch.tensor(loss))
            tracker.next_epoch()
            with torch.no_grad():
                model.weight.fill_(float(epoch))
            ck.maybe_save(model, epoch=epoch, tracker=tracker)

        best = ck.load('best')
        assert best['epoch'] == 2  # loss=1.0 was the best
        assert best['state_dict']['weight'][0, 0].item() == 2.0


if __name__ == '__main__':
    sys.exit(pytest.main([__file__]))
