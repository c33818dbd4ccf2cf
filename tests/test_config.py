"""Config system (OmegaConf-equivalent surface the pipeline needs)."""

import sys

import pytest

from dmlcloud_amd.config import Config


class TestConfig:
    def test_create_and_access(self):
        cfg = Config.create({'a': 1, 'b': {'c': 2}})
        assert cfg.a == 1
        assert cfg.b.c == 2
        assert cfg['b']['c'] == 2

    def test_create_none(self):
        cfg = Config.create(None)
        assert len(cfg) == 0

    def test_create_passthrough(self):
        cfg = Config.create({'x': 1})
        assert Config.create(cfg) is cfg

    def test_set_nested(self):
        cfg = Config()
        cfg.model = {'layers': 4}
        assert cfg.model.layers == 4
        cfg.model.layers = 8
        assert cfg['model']['layers'] == 8

    def test_missing_raises(self):
        cfg = Config()
        with pytest.raises(AttributeError):
            _ = cfg.nope
        with pytest.raises(KeyError):
            _ = cfg['nope']

    def test_yaml_roundtrip(self, tmp_path):
        cfg = Config.create({'a': 1, 'b': {'c': [1, 2, 3]}, 's': 'text'})
        path = tmp_path / 'c.yaml'
        cfg.save(path)
        loaded = Config.load(path)
        assert loaded == cfg

    def test_interpolation(self):
        cfg = Config.create({'root': '/data', 'path': '${root}/train', 'n': 4, 'alias': '${n}'})
        resolved = cfg.to_container(resolve=True)
        assert resolved['path'] == '/data/train'
        assert resolved['alias'] == 4  # whole-string interp preserves type

    def test_interpolation_nested(self):
        cfg = Config.create({'a': {'b': 7}, 'c': '${a.b}'})
        assert cfg.to_container(resolve=True)['c'] == 7

    def test_merge(self):
        a = Config.create({'x': 1, 'sub': {'y': 2, 'z': 3}})
        b = a.merge({'sub': {'y': 20}, 'w': 4})
        assert b.sub.y == 20
        assert b.sub.z == 3
        assert b.w == 4
        assert a.sub.y == 2  # original untouched

    def test_to_container_is_plain(self):
        cfg = Config.create({'a': {'b': 1}})
        container = cfg.to_container()
        assert isinstance(container, dict)
        assert isinstance(container['a'], dict)


class TestProgressTable:
    def test_render(self):
        import io

        from dmlcloud_amd.utils.table import ProgressTable

        buf = io.StringIO()
        table = ProgressTable(file=buf)
        table.add_column('Epoch')
        table.add_column('Loss')
        table['Epoch'] = 1
        table.update('Loss', 0.123456)
        table.next_row()
        table.update('Epoch', 2)
        table.update('Loss', 0.1)
        table.next_row()
        table.close()
        out = buf.getvalue()
        assert 'Epoch' in out and 'Loss' in out
        assert '0.1' in out
        assert out.count('\n') >= 5  # header box + 2 rows + bottom


if __name__ == '__main__':
    sys.exit(pytest.main([__file__]))
