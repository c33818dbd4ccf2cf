"""Introspection of the *user's* project (not this library).

Used to record the git state of the training script's repository for
reproducibility. Capability parity with reference
dmlcloud/util/project.py:7-77 (script_path / project_dir / run_in_project),
reimplemented.
"""

import subprocess
import sys
from pathlib import Path
from typing import Optional


def script_path() -> Optional[Path]:
    """Best-effort path of the entry-point script that is currently running.

    Returns None for interactive sessions.
    """
    main = sys.modules.get('__main__')
    if main is None:
        return None
    path = getattr(main, '__file__', None)
    if path is None:
        return None
    return Path(path).resolve()


def project_dir() -> Optional[Path]:
    """Walk up from the entry-point script past any package __init__.py files.

    Heuristic for the root directory of the user's project.
    """
    path = script_path()
    if path is None:
        return None
    directory = path.parent
    while (directory / '__init__.py').exists() and directory.parent != directory:
        directory = directory.parent
    return directory


def run_in_project(cmd):
    """Run a subprocess with cwd set to the user's project directory."""
    cwd = project_dir() or Path.cwd()
    return subprocess.run(cmd, cwd=cwd, capture_output=True)
