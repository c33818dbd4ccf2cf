"""bench.py driver contract: runs standalone, emits one valid JSON line."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench(extra):
    proc = subprocess.run(
        [sys.executable, 'bench.py', '--steps', '2', '--warmup', '1', '--batch-size', '64'] + extra,
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=420,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [ln for ln in proc.stdout.splitlines() if ln.startswith('{')]
    assert len(lines) == 1, proc.stdout
    return json.loads(lines[0])


@pytest.mark.parametrize('model', ['mnist'])
def test_bench_json_contract(model):
    result = _run_bench(['--model', model])
    for key in (
        'metric',
        'value',
        'unit',
        'n_gpus',
        'steps',
        'warmup',
        'ms_per_step',
        'higher_is_better',
        'scaling',
        'vs_baseline',
        'dtype',
        'data',
        'config',
    ):
        assert key in result, key
    assert result['n_gpus'] == 1
    assert result['steps'] == 2
    assert result['data'] == 'synthetic'
    assert result['value'] > 0
    assert result['config']['parallelism'] == 'dp1'
    assert result['scaling'] == 'weak'


def test_bench_ddp_impl():
    result = _run_bench(['--impl', 'ddp'])
    assert result['config']['impl'] == 'ddp'


def test_bench_metric_stress_mode():
    """BASELINE config #5 plumbing: N per-step reducers tracked and
    reported in the JSON config."""
    result = _run_bench(['--metric-stress', '8'])
    assert result['config']['metric_stress'] == 8


def _free_port():
    import socket

    s = socket.socket()
    s.bind(('127.0.0.1', 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.mark.parametrize('impl', ['flat', 'ddp'])
def test_bench_torchrun_w2(impl):
    """The driver's multi-GPU launch contract, exercised at world_size=2 on
    CPU/gloo: torch.distributed.run -> env:// rendezvous -> ONE JSON line
    with n_gpus=2 / dp2."""
    proc = subprocess.run(
        [
            sys.executable,
            '-m',
            'torch.distributed.run',
            '--nnodes=1',
            '--nproc-per-node',
            '2',
            '--master-addr',
            '127.0.0.1',
            '--master-port',
            str(_free_port()),
            'bench.py',
            '--gpus',
            '2',
            '--steps',
            '2',
            '--warmup',
            '1',
            '--batch-size',
            '64',
            '--impl',
            impl,
        ],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=420,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [ln for ln in proc.stdout.splitlines() if ln.startswith('{')]
    assert len(lines) == 1, proc.stdout  # rank 0 only
    result = json.loads(lines[0])
    assert result['n_gpus'] == 2
    assert result['config']['parallelism'] == 'dp2'
    assert result['config']['global_batch'] == 128
    assert result['value'] > 0
