"""Optional third-party module probes for startup diagnostics.

Capability parity with reference dmlcloud/util/thirdparty.py:7-36.
"""

import importlib
import sys
from types import ModuleType
from typing import Optional

ML_MODULES = [
    'torch',
    'torchvision',
    'torchaudio',
    'einops',
    'numpy',
    'pandas',
    'xarray',
    'sklearn',
    'transformers',
]


def is_imported(name: str) -> bool:
    return name in sys.modules


def try_import(name: str) -> Optional[ModuleType]:
    try:
        return importlib.import_module(name)
    except ImportError:
        return None


def try_get_version(name: str) -> Optional[str]:
    module = try_import(name)
    if module is not None:
        return str(getattr(module, '__version__', 'unknown'))
    return None
