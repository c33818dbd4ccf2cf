"""SLURM environment accessors.

Capability parity with the reference's SLURM glue (reference:
dmlcloud/util/slurm.py:4-13): plain env-var reads used by checkpoint
rediscovery and diagnostics.
"""

import os


def slurm_job_id():
    return os.environ.get('SLURM_JOB_ID')


def slurm_step_id():
    return os.environ.get('SLURM_STEP_ID')


def slurm_available() -> bool:
    return slurm_job_id() is not None
