"""SLURM environment introspection.

Capability parity with the reference's SLURM glue (reference
dmlcloud/util/slurm.py:4-13) — env-var reads used by checkpoint
rediscovery and startup diagnostics — plus the handful of extra fields
the diagnostics block reports.
"""

import os
from typing import Dict, Optional

__all__ = ['slurm_job_id', 'slurm_step_id', 'slurm_available', 'slurm_summary']

# diagnostics-relevant SLURM variables beyond the job/step ids
_REPORTED_VARS = (
    'SLURM_STEP_NODELIST',
    'SLURM_TASKS_PER_NODE',
    'SLURM_STEP_GPUS',
    'SLURM_GPUS_ON_NODE',
    'SLURM_CPUS_PER_TASK',
)


def _env(name: str) -> Optional[str]:
    return os.environ.get(name)


def slurm_job_id() -> Optional[str]:
    return _env('SLURM_JOB_ID')


def slurm_step_id() -> Optional[str]:
    return _env('SLURM_STEP_ID')


def slurm_available() -> bool:
    """True when running inside a SLURM allocation."""
    return slurm_job_id() is not None


def slurm_summary() -> Dict[str, Optional[str]]:
    """The SLURM facts worth logging at startup (empty dict off-SLURM)."""
    if not slurm_available():
        return {}
    summary = {'SLURM_JOB_ID': slurm_job_id(), 'SLURM_STEP_ID': slurm_step_id()}
    for name in _REPORTED_VARS:
        summary[name] = _env(name)
    return summary
