"""Record git state of the user's project for experiment reproducibility.

Capability parity with reference dmlcloud/util/git.py:4-14.
"""

from .project import run_in_project


def git_hash(short: bool = False) -> str:
    args = ['git', 'rev-parse', '--short', 'HEAD'] if short else ['git', 'rev-parse', 'HEAD']
    process = run_in_project(args)
    return process.stdout.decode('utf-8').strip()


def git_diff() -> str:
    process = run_in_project(['git', 'diff', '-U0', '--no-color', 'HEAD'])
    return process.stdout.decode('utf-8').strip()
