import sys

import pytest


def test_import():
    import dmlcloud_amd

    assert dmlcloud_amd.__version__
    assert hasattr(dmlcloud_amd, 'Stage')
    assert hasattr(dmlcloud_amd, 'TrainValStage')
    assert hasattr(dmlcloud_amd, 'TrainingPipeline')


def test_version_sane():
    import dmlcloud_amd

    parts = dmlcloud_amd.__version__.split('.')
    assert len(parts) >= 2
    assert all(p.isdigit() for p in parts[:2])


def test_submodules():
    import dmlcloud_amd.checkpoint  # noqa: F401
    import dmlcloud_amd.data  # noqa: F401
    import dmlcloud_amd.metrics  # noqa: F401
    import dmlcloud_amd.models  # noqa: F401
    import dmlcloud_amd.ops  # noqa: F401
    import dmlcloud_amd.parallel  # noqa: F401
    import dmlcloud_amd.utils  # noqa: F401


def test_extension_importable():
    """The gfx950 extension must be importable here (built for CPU-side
    import; kernels only run on a GPU box)."""
    from dmlcloud_amd import ops

    assert ops.is_available(), f'native extension missing: {ops._EXT_ERROR}'


if __name__ == '__main__':
    sys.exit(pytest.main([__file__]))
