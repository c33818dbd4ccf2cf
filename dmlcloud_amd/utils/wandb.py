"""Lazy wandb integration (optional dependency).

Capability parity with reference dmlcloud/util/wandb.py:5-30. wandb is not
required to import or use the framework; everything here degrades to no-ops
or raises only when actually used without wandb installed.
"""

import os
import sys


class WandbModuleWrapper:
    """Defers the (slow) wandb import until first attribute access."""

    def __getattr__(self, name):
        import wandb as _wandb

        return getattr(_wandb, name)

    def __setattr__(self, name, value):
        import wandb as _wandb

        setattr(_wandb, name, value)


wandb = WandbModuleWrapper()


def wandb_available() -> bool:
    try:
        import wandb as _  # noqa: F401

        return True
    except ImportError:
        return False


def wandb_set_startup_timeout(seconds: int):
    assert isinstance(seconds, int)
    os.environ['WANDB__SERVICE_WAIT'] = f'{seconds}'


def wandb_is_imported() -> bool:
    return 'wandb' in sys.modules


def wandb_is_initialized() -> bool:
    return wandb.run is not None
