"""Seeding and determinism toggles.

Capability parity with reference dmlcloud/util/seed.py:7-15, plus
ROCm-specific determinism knobs (MIOpen find-mode).
"""

import os
import random

import numpy as np
import torch


def seed_all(seed: int):
    """Seed torch, numpy and the stdlib RNG."""
    torch.manual_seed(seed)
    np.random.seed(seed)
    random.seed(seed)


def enable_determinism():
    """Force deterministic kernels.

    On ROCm this additionally pins MIOpen to the default find mode so conv
    algorithm selection (and therefore numerics) is stable across runs.
    """
    torch.backends.cudnn.benchmark = False
    torch.use_deterministic_algorithms(True)
    # MIOpen: immediate mode avoids on-disk find-db nondeterminism across boxes
    os.environ.setdefault('MIOPEN_FIND_MODE', '1')
