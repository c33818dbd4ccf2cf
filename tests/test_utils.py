"""Utility-belt coverage: argparse, tcp, slurm, git, thirdparty, seed,
logging handlers, wandb wrapper."""

import argparse
import enum
import logging
import socket
import sys

import pytest
import torch

from dmlcloud_amd.utils.argparse import EnumAction
from dmlcloud_amd.utils.logging import DevNullIO, IORedirector, add_log_handlers, flush_log_handlers
from dmlcloud_amd.utils.seed import seed_all
from dmlcloud_amd.utils.slurm import slurm_available, slurm_job_id
from dmlcloud_amd.utils.tcp import find_free_port, get_local_ips
from dmlcloud_amd.utils.thirdparty import is_imported, try_get_version, try_import


class Color(enum.Enum):
    RED = 'red'
    BLUE = 'blue'


class TestEnumAction:
    def test_parses(self):
        parser = argparse.ArgumentParser()
        parser.add_argument('--color', type=Color, action=EnumAction)
        args = parser.parse_args(['--color', 'red'])
        assert args.color is Color.RED

    def test_rejects_invalid(self):
        parser = argparse.ArgumentParser()
        parser.add_argument('--color', type=Color, action=EnumAction)
        with pytest.raises(SystemExit):
            parser.parse_args(['--color', 'green'])

    def test_requires_enum(self):
        parser = argparse.ArgumentParser()
        with pytest.raises(ValueError):
            parser.add_argument('--x', action=EnumAction)
        with pytest.raises(TypeError):
            parser.add_argument('--y', type=int, action=EnumAction)

    def test_nargs_list(self):
        parser = argparse.ArgumentParser()
        parser.add_argument('--colors', type=Color, action=EnumAction, nargs='+')
        args = parser.parse_args(['--colors', 'red', 'blue'])
        assert args.colors == [Color.RED, Color.BLUE]


class TestTcp:
    def test_free_port_bindable(self):
        port = find_free_port()
        with socket.socket() as s:
            s.bind(('', port))

    def test_local_ips(self):
        ips = get_local_ips()
        assert isinstance(ips, list) and len(ips) >= 1


class TestSlurm:
    def test_absent(self, monkeypatch):
        monkeypatch.delenv('SLURM_JOB_ID', raising=False)
        assert slurm_job_id() is None
        assert not slurm_available()

    def test_present(self, monkeypatch):
        monkeypatch.setenv('SLURM_JOB_ID', '77')
        assert slurm_job_id() == '77'
        assert slurm_available()


class TestThirdparty:
    def test_torch_probe(self):
        assert is_imported('torch')
        assert try_get_version('torch') == torch.__version__
        assert try_import('definitely_not_a_module_xyz') is None
        assert try_get_version('definitely_not_a_module_xyz') is None

    def test_installed_versions(self):
        from dmlcloud_amd.utils.thirdparty import installed_versions

        versions = installed_versions()
        assert 'torch' in versions and 'numpy' in versions
        assert all(isinstance(v, str) for v in versions.values())


class TestSeed:
    def test_seed_reproducible(self):
        seed_all(123)
        a = torch.randn(4)
        seed_all(123)
        b = torch.randn(4)
        torch.testing.assert_close(a, b)

    def test_seed_for_rank_diverges_per_rank(self):
        from dmlcloud_amd.utils.seed import seed_for_rank

        s0 = seed_for_rank(7, rank_=0)
        a = torch.randn(4)
        s1 = seed_for_rank(7, rank_=1)
        b = torch.randn(4)
        assert s0 != s1
        assert not torch.equal(a, b)
        # deterministic per (seed, rank)
        assert seed_for_rank(7, rank_=0) == s0
        torch.testing.assert_close(torch.randn(4), a)


class TestSlurmSummary:
    def test_off_slurm_empty(self, monkeypatch):
        monkeypatch.delenv('SLURM_JOB_ID', raising=False)
        from dmlcloud_amd.utils.slurm import slurm_summary

        assert slurm_summary() == {}

    def test_on_slurm_reports(self, monkeypatch):
        monkeypatch.setenv('SLURM_JOB_ID', '42')
        monkeypatch.setenv('SLURM_CPUS_PER_TASK', '16')
        from dmlcloud_amd.utils.slurm import slurm_summary

        facts = slurm_summary()
        assert facts['SLURM_JOB_ID'] == '42'
        assert facts['SLURM_CPUS_PER_TASK'] == '16'


class TestLogging:
    def test_devnull(self):
        DevNullIO().write('anything')

    def test_io_redirector_context_and_idempotence(self, tmp_path):
        log = tmp_path / 'log.txt'
        red = IORedirector(log)
        with red:
            red.install()  # second install is a no-op
            print('inside-context')
        assert not red.active
        assert 'inside-context' in log.read_text()
        # uninstall after uninstall is also a no-op
        red.uninstall()

    def test_io_redirector(self, tmp_path):
        log = tmp_path / 'log.txt'
        log.touch()
        red = IORedirector(log)
        red.install()
        try:
            print('tee-me')
        finally:
            red.uninstall()
        assert 'tee-me' in log.read_text()
        print('not-teed')
        assert 'not-teed' not in log.read_text()

    def test_handlers(self, torch_distributed):
        logger = logging.getLogger('dml-test-logger')
        logger.handlers.clear()
        logger.propagate = False  # hasHandlers() must not see pytest's root handlers
        add_log_handlers(logger)
        assert len(logger.handlers) == 2
        flush_log_handlers(logger)
        logger.handlers.clear()


class TestWandbWrapper:
    def test_lazy_no_import(self):
        from dmlcloud_amd.utils import wandb as w

        assert not w.wandb_is_imported() or 'wandb' in sys.modules
        assert w.wandb_available() in (True, False)

    def test_startup_timeout(self, monkeypatch):
        from dmlcloud_amd.utils.wandb import wandb_set_startup_timeout

        wandb_set_startup_timeout(42)
        import os

        assert os.environ['WANDB__SERVICE_WAIT'] == '42'


class TestGitProject:
    def test_git_hash_runs(self):
        from dmlcloud_amd.utils.git import git_hash

        h = git_hash()
        assert isinstance(h, str)


class TestDiagnostics:
    def test_general_diagnostics(self, torch_distributed):
        from dmlcloud_amd.utils.logging import experiment_header, general_diagnostics

        diag = general_diagnostics()
        assert 'VERSIONS' in diag
        assert 'hip' in diag
        from datetime import datetime

        header = experiment_header('exp', None, datetime.now())
        assert 'exp' in header


if __name__ == '__main__':
    sys.exit(pytest.main([__file__]))
