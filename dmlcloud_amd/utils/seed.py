"""Seeding and determinism toggles.

Capability parity with reference dmlcloud/util/seed.py:7-15, extended
with rank-aware seeding and the ROCm-specific determinism knobs
(MIOpen find-mode, hipBLASLt workspace pinning).
"""

import os
import random

import numpy as np
import torch

__all__ = ['seed_all', 'seed_for_rank', 'enable_determinism']


def seed_all(seed: int):
    """Seed every RNG the training stack draws from: torch (CPU + all
    GPUs), numpy's legacy global generator, and the stdlib."""
    torch.manual_seed(seed)
    np.random.seed(seed % (2**32))
    random.seed(seed)


def seed_for_rank(seed: int, rank_: int = None) -> int:
    """Derive a per-rank seed and apply it.

    Data-parallel ranks usually want DIFFERENT streams (dropout,
    augmentation) derived from ONE experiment seed; sharding decisions
    stay rank-independent because they take explicit seeds. Returns the
    derived seed so it can be logged.
    """
    if rank_ is None:
        from ..parallel.distributed import rank

        rank_ = rank() or 0
    derived = (seed * 0x9E3779B1 + rank_) % (2**31)
    seed_all(derived)
    return derived


def enable_determinism():
    """Force deterministic kernels.

    - disables MIOpen benchmark/find nondeterminism (algorithm choice
      varies with measured timings run-to-run otherwise),
    - turns on torch's deterministic-algorithms enforcement,
    - pins the hipBLASLt workspace config so GEMM splits stay fixed.

    Call BEFORE the first conv/GEMM executes.
    """
    torch.backends.cudnn.benchmark = False
    torch.use_deterministic_algorithms(True)
    # MIOpen: normal find mode reads the deterministic find-db instead of
    # re-timing candidates; immediate-mode heuristics are also stable
    os.environ.setdefault('MIOPEN_FIND_MODE', '1')
    # rocBLAS/hipBLASLt: required by torch for deterministic GEMM workspaces
    os.environ.setdefault('CUBLAS_WORKSPACE_CONFIG', ':4096:8')
