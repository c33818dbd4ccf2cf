"""Bootstrap-ladder tests: launcher detection and the SLURM init path
(simulated via env vars at world_size=1 — the mechanism SURVEY.md §3.1
documents; the reference leaves these paths untested)."""

import sys

import pytest
import torch.distributed as dist

from dmlcloud_amd.parallel.distributed import (
    deinitialize_torch_distributed,
    has_environment,
    has_mpi,
    has_slurm,
    init_process_group_slurm,
    local_rank,
    local_world_size,
    rank,
    world_size,
)
from dmlcloud_amd.utils.tcp import find_free_port

_SLURM_WORLD1 = {
    'SLURM_PROCID': '0',
    'SLURM_NTASKS': '1',
    'SLURM_LOCALID': '0',
    'SLURM_STEP_TASKS_PER_NODE': '1',
    'SLURM_NODEID': '0',
    'SLURM_SRUN_COMM_HOST': '127.0.0.1',
}


class TestLauncherDetection:
    def test_env_probe(self, monkeypatch):
        monkeypatch.delenv('MASTER_PORT', raising=False)
        assert not has_environment()
        monkeypatch.setenv('MASTER_PORT', '12345')
        assert has_environment()

    def test_slurm_probe(self, monkeypatch):
        monkeypatch.delenv('SLURM_PROCID', raising=False)
        assert not has_slurm()
        monkeypatch.setenv('SLURM_PROCID', '0')
        assert has_slurm()

    def test_mpi_probe_returns_bool(self):
        assert has_mpi() in (True, False)


class TestSlurmInitPath:
    def test_world1_rendezvous(self, monkeypatch):
        """The SLURM path does a real tcp:// rendezvous and fills the
        worker topology from the SLURM_* variables."""
        for key, val in _SLURM_WORLD1.items():
            monkeypatch.setenv(key, val)
        init_process_group_slurm(port=find_free_port(), backend='gloo')
        try:
            assert dist.is_initialized()
            assert dist.get_world_size() == 1
            assert rank() == 0
            assert world_size() == 1
            assert local_rank() == 0
            assert local_world_size() == 1
        finally:
            deinitialize_torch_distributed()
        assert rank() is None  # topology reset on teardown

    def test_tasks_per_node_list_syntax(self, monkeypatch):
        """SLURM_STEP_TASKS_PER_NODE can be '1(x2)' or '1,2'; the leading
        count is what local_world_size reports."""
        for key, val in _SLURM_WORLD1.items():
            monkeypatch.setenv(key, val)
        monkeypatch.setenv('SLURM_STEP_TASKS_PER_NODE', '1(x2)')
        init_process_group_slurm(port=find_free_port(), backend='gloo')
        try:
            assert local_world_size() == 1
        finally:
            deinitialize_torch_distributed()


if __name__ == '__main__':
    sys.exit(pytest.main([__file__]))
