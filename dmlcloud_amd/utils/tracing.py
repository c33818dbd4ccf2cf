"""roctx range annotation for rocprofv3 timelines.

The reference has only hand-rolled wall timers (reference
dmlcloud/stage.py:299-314). Here every Stage epoch and
train step can be wrapped in roctx ranges so `rocprofv3 --sys-trace`
attributes GPU time to pipeline phases. torch.cuda.nvtx maps to
roctracer/roctx on ROCm builds; everything degrades to a no-op when
ranges are unavailable or tracing is disabled.
"""

import os
from contextlib import contextmanager

import torch

_ENABLED = os.environ.get('DMLCLOUD_ROCTX', '0') not in ('0', '', 'false', 'False')


def tracing_enabled() -> bool:
    return _ENABLED


def enable_tracing(on: bool = True):
    global _ENABLED
    _ENABLED = on


def range_push(name: str):
    if _ENABLED:
        try:
            torch.cuda.nvtx.range_push(name)
        except Exception:
            pass


def range_pop():
    if _ENABLED:
        try:
            torch.cuda.nvtx.range_pop()
        except Exception:
            pass


@contextmanager
def roctx_range(name: str):
    """Context manager marking a roctx range (no-op unless enabled)."""
    range_push(name)
    try:
        yield
    finally:
        range_pop()
