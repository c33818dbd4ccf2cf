"""hipGraph-captured training steps.

MI355X-idiomatic replacement for a tracing compiler: the training step of
a static model is a fixed kernel sequence on fixed pointers (see
parallel/flat.py), so it is captured once into a hipGraph
(torch.cuda.CUDAGraph == hipGraph on ROCm) and replayed with a single
host-side launch per step. This collapses the per-step launch overhead
that dominates small/medium models (the headline MNIST-CNN benchmark is
launch-bound in eager mode).

RCCL collectives are capturable, so the flat all-reduce of
FlatReplica.grad_sync() rides inside the graph.
"""

import logging
from typing import Callable, Optional

import torch
import torch.distributed as dist

logger = logging.getLogger('dmlcloud_amd')


def _capture_unanimous(captured: bool) -> bool:
    """Vote across ranks on capture success.

    The decision to replay (including the validation replay, which
    EXECUTES any captured collectives) must be collective-consistent:
    a rank replaying an all-reduce while another rank skips it deadlocks
    the job. If any rank failed to capture, every rank runs eager — the
    per-step collective sequence is identical either way, but the extra
    validation replay is not.
    """
    if not (dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1):
        return captured
    votes = [None] * dist.get_world_size()
    dist.all_gather_object(votes, captured)
    return all(votes)


class GraphedStep:
    """Capture `step_fn` (no arguments; reads/writes static tensors) into a
    hipGraph after `warmup` eager iterations on a side stream.

    Usage:
        gs = GraphedStep(run_step, warmup=3)
        gs.initialize()          # warmup + capture (falls back to eager on error)
        ...
        gs()                     # replay (or eager fallback)
    """

    def __init__(
        self, step_fn: Callable[[], None], warmup: int = 3, enabled: bool = True, validate: bool = False
    ):
        """validate=True replays the captured graph once (and synchronizes)
        inside initialize() so replay-time failures — e.g. an RCCL
        collective that captured but cannot replay at this world size —
        also trigger the eager fallback. The validation replay advances
        model/optimizer state by one step, so enable it only when
        initialize() runs inside a warmup phase (the benchmark does)."""
        self.step_fn = step_fn
        self.warmup = warmup
        self.enabled = enabled and torch.cuda.is_available()
        self.validate = validate
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self._initialized = False

    def initialize(self):
        if self._initialized:
            return
        self._initialized = True
        if not self.enabled:
            return
        graph = None
        try:
            stream = torch.cuda.Stream()
            stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                for _ in range(self.warmup):
                    self.step_fn()
            torch.cuda.current_stream().wait_stream(stream)
            torch.cuda.synchronize()

            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                self.step_fn()
        except Exception as e:  # pragma: no cover - depends on runtime support
            logger.warning(f'hipGraph capture failed ({e!r}); falling back to eager stepping')
            graph = None

        # all ranks must agree before anything replays (see _capture_unanimous)
        if not _capture_unanimous(graph is not None):
            if graph is not None:
                logger.warning('hipGraph captured here but not on every rank; using eager everywhere')
            self.graph = None
            return

        if graph is not None and self.validate:
            try:
                graph.replay()
                torch.cuda.synchronize()
            except Exception as e:  # pragma: no cover
                logger.warning(f'hipGraph validation replay failed ({e!r}); eager fallback')
                graph = None
        self.graph = graph

    @property
    def captured(self) -> bool:
        return self.graph is not None

    def __call__(self):
        if not self._initialized:
            self.initialize()
        if self.graph is not None:
            self.graph.replay()
        else:
            self.step_fn()
