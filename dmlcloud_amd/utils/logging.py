"""Logging, stdio capture, and the startup diagnostics report.

Three jobs (capability parity with reference dmlcloud/util/logging.py:
18-173, reshaped for the ROCm stack):

1. ``IORedirector`` — capture everything written to stdout/stderr into
   the run's ``log.txt`` while still reaching the terminal. Implemented
   as a single tee proxy class used for both streams.
2. Rank-aware handler setup for the ``'dmlcloud'`` logger: rank 0 is
   chatty (INFO), other ranks only surface problems (WARNING); records
   below WARNING go to stdout, the rest to stderr.
3. ``general_diagnostics`` — a provenance block logged at startup:
   process identity, git state, distributed backend, the ROCm/HIP
   stack (``amd-smi`` / ``rocm-smi`` / ``torch.version.hip`` — the
   CUDA probes of the reference have no ROCm analog and are replaced,
   reference logging.py:146,155), package versions, and SLURM context.
"""

import io
import logging
import os
import subprocess
import sys
from datetime import datetime
from pathlib import Path
from typing import List, Optional, TextIO

import torch
import torch.distributed as dist

from .git import git_hash
from .slurm import slurm_summary
from .thirdparty import installed_versions

__all__ = [
    'IORedirector',
    'DevNullIO',
    'add_log_handlers',
    'flush_log_handlers',
    'experiment_header',
    'general_diagnostics',
]


class _Tee:
    """File-like proxy that writes to a live terminal stream and a file."""

    def __init__(self, terminal: TextIO, sink: TextIO):
        self._terminal = terminal
        self._sink = sink

    def write(self, text):
        self._sink.write(text)
        return self._terminal.write(text)

    def flush(self):
        self._sink.flush()
        self._terminal.flush()

    def __getattr__(self, attr):
        # isatty, encoding, fileno, ... — delegate to the real terminal
        return getattr(self._terminal, attr)


class IORedirector:
    """Mirror stdout and stderr into ``log_file`` (append mode).

    ``install()`` swaps ``sys.stdout``/``sys.stderr`` for tee proxies;
    ``uninstall()`` restores the originals and closes the file. Also
    usable as a context manager. Installing twice is a no-op.
    """

    def __init__(self, log_file: Path):
        self.path = Path(log_file)
        self._log: Optional[TextIO] = None
        self._saved = None

    @property
    def active(self) -> bool:
        return self._log is not None

    def install(self):
        if self.active:
            return
        self._log = open(self.path, 'a', buffering=1)
        self._saved = (sys.stdout, sys.stderr)
        for stream in self._saved:
            stream.flush()
        sys.stdout = _Tee(self._saved[0], self._log)
        sys.stderr = _Tee(self._saved[1], self._log)

    def uninstall(self):
        if not self.active:
            return
        sys.stdout.flush()
        sys.stderr.flush()
        sys.stdout, sys.stderr = self._saved
        self._log.close()
        self._log = None
        self._saved = None

    def __enter__(self):
        self.install()
        return self

    def __exit__(self, *exc_info):
        self.uninstall()

    # kept for callers that used the attribute names of the old API
    @property
    def file(self):
        return self._log


class DevNullIO(io.TextIOBase):
    """Write sink that discards everything (non-root progress tables)."""

    def write(self, text):
        return len(text) if text else 0

    def writable(self):
        return True


def _is_root_rank() -> bool:
    return not dist.is_initialized() or dist.get_rank() == 0


def add_log_handlers(logger: logging.Logger):
    """Attach the rank-aware stdout/stderr handler pair (idempotent).

    Rank 0: INFO and up. Other ranks: WARNING and up. Severity below
    WARNING renders to stdout, WARNING+ to stderr — so redirected logs
    interleave correctly with print() output.
    """
    if logger.hasHandlers():
        return
    logger.setLevel(logging.INFO if _is_root_rank() else logging.WARNING)

    info_handler = logging.StreamHandler(sys.stdout)
    info_handler.setLevel(logging.DEBUG)
    info_handler.addFilter(lambda rec: rec.levelno < logging.WARNING)
    logger.addHandler(info_handler)

    problem_handler = logging.StreamHandler(sys.stderr)
    problem_handler.setLevel(logging.WARNING)
    logger.addHandler(problem_handler)


def flush_log_handlers(logger: logging.Logger):
    for handler in logger.handlers:
        handler.flush()


def experiment_header(name, checkpoint_dir, date: datetime) -> str:
    """Banner logged when a pipeline starts."""
    world = dist.get_world_size() if dist.is_initialized() else 1
    lines = [
        f'...............  Experiment: {name or "N/A"}  ...............',
        f'- Date: {date}',
        f'- Checkpoint Dir: {checkpoint_dir or "N/A"}',
        f'- Training on {world} GPUs',
    ]
    return '\n'.join(lines) + '\n'


# ------------------------------------------------------------- diagnostics


def _read_rocm_release() -> Optional[str]:
    for candidate in ('/opt/rocm/.info/version', '/opt/rocm/.info/version-dev'):
        try:
            return Path(candidate).read_text().strip()
        except OSError:
            continue
    return None


def _probe_gpus() -> List[str]:
    """GPU inventory via the AMD SMI tools, torch as a fallback."""
    smi_commands = (
        ['amd-smi', 'list', '--csv'],
        ['rocm-smi', '--showproductname'],
    )
    for command in smi_commands:
        try:
            result = subprocess.run(
                command, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, timeout=30
            )
        except (FileNotFoundError, subprocess.TimeoutExpired):
            continue
        if result.returncode == 0:
            return [ln for ln in result.stdout.decode().splitlines() if ln.strip()]
    return [torch.cuda.get_device_name(i) for i in range(torch.cuda.device_count())]


def _bullet_block(title: str, entries) -> str:
    body = ''.join(f'    - {key}: {val}\n' for key, val in entries)
    return f'* {title}:\n{body}'


def general_diagnostics() -> str:
    """Multi-line provenance report for the start-of-run log."""
    import dmlcloud_amd

    general = [
        ('argv', sys.argv),
        ('cwd', Path.cwd()),
        ('host (root)', os.environ.get('HOSTNAME')),
        ('user', os.environ.get('USER')),
        ('git-hash', git_hash()),
        ('conda-env', os.environ.get('CONDA_DEFAULT_ENV', 'N/A')),
        ('sys-prefix', sys.prefix),
        ('backend', dist.get_backend() if dist.is_initialized() else 'N/A'),
        ('gpu available', torch.cuda.is_available()),
    ]
    report = _bullet_block('GENERAL', general)

    if torch.cuda.is_available():
        report += _bullet_block('GPUs (root)', [('gpu', g) for g in _probe_gpus()]).replace(
            '- gpu: ', '- '
        )

    versions = [
        ('python', sys.version),
        ('dmlcloud_amd', dmlcloud_amd.__version__),
        ('hip', torch.version.hip),
        ('rocm', _read_rocm_release() or 'N/A'),
    ]
    versions += sorted(installed_versions().items())
    report += _bullet_block('VERSIONS', versions)

    slurm_facts = slurm_summary()
    if slurm_facts:
        report += _bullet_block('SLURM', slurm_facts.items())

    return report.rstrip('\n')
