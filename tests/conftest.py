import pytest
import torch

from dmlcloud_amd.parallel import deinitialize_torch_distributed, init_process_group_dummy


def pytest_configure(config):
    config.addinivalue_line('markers', 'gpu: test requires an MI355X GPU (run with -m gpu on a GPU box)')


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason='no GPU available')
    for item in items:
        if 'gpu' in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def torch_distributed():
    """World-size-1 dummy process group (HashStore) — every collective code
    path runs for real."""
    init_process_group_dummy(backend='gloo')
    yield
    deinitialize_torch_distributed()


@pytest.fixture
def torch_distributed_cuda():
    """Dummy group with the default backend (RCCL on a GPU box)."""
    init_process_group_dummy()
    yield
    deinitialize_torch_distributed()
