"""Optional-dependency probes for the startup diagnostics block.

`general_diagnostics` wants version strings for whatever ML packages
happen to be installed, without importing anything heavyweight that the
user did not already import (capability parity with reference
dmlcloud/util/thirdparty.py:7-36). Version lookup therefore prefers
importlib.metadata (no import side effects) and only falls back to an
actual import for packages whose metadata is unavailable.
"""

import importlib
import importlib.metadata
import sys
from types import ModuleType
from typing import Dict, Optional

__all__ = ['ML_MODULES', 'is_imported', 'try_import', 'try_get_version', 'installed_versions']

# Packages worth reporting in diagnostics, roughly by relevance on a
# ROCm training box.
ML_MODULES = [
    'torch',
    'numpy',
    'einops',
    'transformers',
    'datasets',
    'safetensors',
    'xarray',
    'pandas',
    'sklearn',
    'torchvision',
    'torchaudio',
]

# distribution names that differ from the import name
_DIST_NAMES = {'sklearn': 'scikit-learn'}


def is_imported(name: str) -> bool:
    """True if the module is already loaded in this interpreter."""
    return name in sys.modules


def try_import(name: str) -> Optional[ModuleType]:
    """Import and return the module, or None if that fails for any reason."""
    try:
        return importlib.import_module(name)
    except Exception:
        return None


def try_get_version(name: str) -> Optional[str]:
    """Best-effort version string for an installed package, else None.

    Order: installed distribution metadata (cheap, no import), then the
    module's ``__version__`` if the module is importable.
    """
    try:
        return importlib.metadata.version(_DIST_NAMES.get(name, name))
    except importlib.metadata.PackageNotFoundError:
        pass
    except Exception:
        pass
    module = try_import(name)
    if module is None:
        return None
    return str(getattr(module, '__version__', 'unknown'))


def installed_versions() -> Dict[str, str]:
    """{module: version} for every ML_MODULES entry that is installed."""
    found = {}
    for name in ML_MODULES:
        version = try_get_version(name)
        if version is not None:
            found[name] = version
    return found
