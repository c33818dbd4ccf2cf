"""dmlcloud_amd: MI355X-native distributed training pipeline framework.

A ground-up AMD CDNA4 implementation of the capabilities of
sehoffmann/dmlcloud v0.3.3 (reference mounted at /root/reference):
TrainingPipeline / Stage / TrainValStage experiment orchestration, a
distributed metric tracker with gfx950 HIP reduction kernels and fused
RCCL epoch-end collectives, real weight checkpointing with a fused
device pack, deterministic dataset sharding, and RCCL-over-xGMI data
parallelism (torch DDP tuned for the 7-link topology, plus a flat-buffer
replica fast path with fused optimizers and hipGraph-captured steps).
"""

from .config import Config  # noqa: F401
from .metrics import MetricReducer, MetricTracker, Reduction  # noqa: F401
from .pipeline import TrainingPipeline  # noqa: F401
from .stage import Stage, TrainValStage  # noqa: F401

__version__ = '0.1.0'

__all__ = [
    'Stage',
    'TrainValStage',
    'TrainingPipeline',
    'MetricTracker',
    'MetricReducer',
    'Reduction',
    'Config',
]
