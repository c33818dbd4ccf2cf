"""Stage engine: epoch state machine with user hook points.

API parity with the reference Stage/TrainValStage (reference
dmlcloud/stage.py:18-341): same hook order (pre_stage -> [pre_epoch ->
run_epoch -> post_epoch]* -> post_stage), same metric namespacing
(train/ val/ misc/), same table semantics and stop_stage behavior.

Internal architecture (this implementation's own):
- table columns are normalized once into ``_ColumnSpec`` records; the
  live table rendering and the per-epoch refresh both work off that list,
- epoch/stage wall-time flows through one ``_Clock`` helper,
- the per-batch hot path is ``TrainValStage.train_batch`` — a single
  method the benchmark driver can call directly (and capture into a
  hipGraph via parallel/graphs.py),
- metric tracking feeds device-resident accumulators (no D2H per batch,
  see metrics.py),
- clip_gradients() computes a fused global-norm clip on flat optimizers
  (single gfx950 kernel chain, no host sync, correct at any world size)
  and falls back to torch.nn.utils for stock optimizers.
"""

import sys
import time
from dataclasses import dataclass
from datetime import datetime, timedelta
from typing import Any, Dict, List, Optional, Union

import torch

from .metrics import MetricTracker, Reduction
from .parallel.distributed import is_root
from .parallel.flat import FlatOptimizer
from .utils.logging import DevNullIO, flush_log_handlers
from .utils.table import ProgressTable
from .utils.tracing import roctx_range

__all__ = ['Stage', 'TrainValStage']


@dataclass
class _ColumnSpec:
    """One progress-table column: display name + the tracked metric that
    feeds it (None = code updates the cell directly, e.g. ETA)."""

    title: str
    metric: Optional[str]
    extra: Dict[str, Any]

    @classmethod
    def parse(cls, raw: Union[str, Dict[str, Any]]) -> '_ColumnSpec':
        if isinstance(raw, str):
            return cls(title=raw, metric=raw, extra={})
        if isinstance(raw, dict):
            missing = {'name', 'metric'} - raw.keys()
            if missing:
                raise ValueError(f'Column dict must contain a "{missing.pop()}" key')
            extra = {k: v for k, v in raw.items() if k not in ('name', 'metric')}
            return cls(title=raw['name'], metric=raw['metric'], extra=extra)
        raise ValueError(f'Invalid column: {raw}. Must be a string or a dict.')


class _Clock:
    """Wall-time bookkeeping for a stage run."""

    def __init__(self):
        self.stage_started: Optional[datetime] = None
        self.stage_stopped: Optional[datetime] = None
        self.epoch_started: Optional[datetime] = None
        self.epoch_stopped: Optional[datetime] = None

    def stage_elapsed(self) -> timedelta:
        return datetime.now() - self.stage_started

    def epoch_seconds(self) -> float:
        return (self.epoch_stopped - self.epoch_started).total_seconds()


class Stage:
    """Hook points: pre_stage, post_stage, pre_epoch, post_epoch."""

    def __init__(self):
        self.pipeline = None  # set by the pipeline on append_stage
        self.max_epochs = None
        self.name = None

        self.current_epoch = 1
        self.metric_prefix = None
        self.table = None
        self.barrier_timeout = None

        self._clock = _Clock()
        self._stop_requested = False
        self._columns: List[_ColumnSpec] = []

    # ------------------------------------------------- pipeline plumbing

    @property
    def tracker(self) -> MetricTracker:
        return self.pipeline.tracker

    @property
    def logger(self):
        return self.pipeline.logger

    @property
    def device(self):
        return self.pipeline.device

    @property
    def config(self):
        return self.pipeline.config

    # timing attributes kept as properties for API compatibility
    @property
    def start_time(self):
        return self._clock.stage_started

    @property
    def stop_time(self):
        return self._clock.stage_stopped

    @property
    def epoch_start_time(self):
        return self._clock.epoch_started

    @property
    def epoch_stop_time(self):
        return self._clock.epoch_stopped

    # ------------------------------------------------------- user surface

    def track_reduce(
        self,
        name: str,
        value: torch.Tensor,
        step: Optional[int] = None,
        reduction: Reduction = Reduction.MEAN,
        dim: Optional[List[int]] = None,
        reduce_globally: bool = True,
        prefixed: bool = True,
    ):
        self.pipeline.track_reduce(self._qualify(name, prefixed), value, step, reduction, dim, reduce_globally)

    def track(self, name: str, value, step: Optional[int] = None, prefixed: bool = True):
        self.pipeline.track(self._qualify(name, prefixed), value, step)

    def _qualify(self, name: str, prefixed: bool) -> str:
        if prefixed and self.metric_prefix:
            return f'{self.metric_prefix}/{name}'
        return name

    def stop_stage(self):
        """Request the epoch loop to end after the current epoch."""
        self._stop_requested = True

    def pre_stage(self):
        """Executed before the stage starts. Register stage-specific
        datasets/models/optimizers here."""

    def post_stage(self):
        """Executed after the stage finishes."""

    def pre_epoch(self):
        """Executed before each epoch."""

    def post_epoch(self):
        """Executed after each epoch, after metrics have been reduced."""

    def run_epoch(self):
        raise NotImplementedError()

    def table_columns(self) -> List[Union[str, Dict[str, Any]]]:
        """Columns for the progress table; strings or dicts with
        'name'/'metric' keys ('metric': None = manually updated)."""
        columns = [
            {'name': 'Epoch', 'metric': 'misc/epoch'},
            {'name': 'Time/Epoch', 'metric': None},
        ]
        if self.max_epochs is not None:
            columns.append({'name': 'ETA', 'metric': None})
        return columns

    # ----------------------------------------------------------- lifecycle

    def run(self):
        """Run until max_epochs or stop_stage()."""
        self._enter_stage()
        while not self._epochs_done():
            self._enter_epoch()
            with roctx_range(f'epoch_{self.current_epoch}'):
                self.run_epoch()
            self._leave_epoch()
        self._leave_stage()

    def _epochs_done(self) -> bool:
        if self._stop_requested:
            return True
        return self.max_epochs is not None and self.current_epoch > self.max_epochs

    def _enter_stage(self):
        self._clock.stage_started = datetime.now()
        self.table = ProgressTable(file=sys.stdout if is_root() else DevNullIO())
        self._columns = [_ColumnSpec.parse(c) for c in self.table_columns()]
        for spec in self._columns:
            self.table.add_column(spec.title, **spec.extra)
        if len(self.pipeline.stages) > 1:
            self.logger.info(f'\n========== STAGE: {self.name} ==========')
        self.pre_stage()
        flush_log_handlers(self.logger)
        self.pipeline.barrier(self.barrier_timeout)

    def _leave_stage(self):
        self.table.close()
        self.post_stage()
        self.pipeline.barrier(self.barrier_timeout)
        self._clock.stage_stopped = datetime.now()
        if len(self.pipeline.stages) > 1:
            elapsed = self._clock.stage_stopped - self._clock.stage_started
            self.logger.info(f'Finished stage in {elapsed}')

    def _enter_epoch(self):
        self._clock.epoch_started = datetime.now()
        self.table['Epoch'] = self.current_epoch
        self.pre_epoch()
        self.pipeline._pre_epoch()

    def _leave_epoch(self):
        self._clock.epoch_stopped = datetime.now()
        self.track(name='misc/epoch', value=self.current_epoch, prefixed=False)
        self.track(name='misc/epoch_time', value=self._clock.epoch_seconds(), prefixed=False)
        self.tracker.next_epoch()
        self.post_epoch()
        self.pipeline._post_epoch()
        self._refresh_table()
        self.current_epoch += 1

    def _refresh_table(self):
        per_epoch = self._clock.stage_elapsed() / self.current_epoch
        self.table.update('Epoch', self.current_epoch)
        self.table.update('Time/Epoch', per_epoch)
        if self.max_epochs is not None:
            self.table.update('ETA', per_epoch * (self.max_epochs - self.current_epoch))
        for spec in self._columns:
            if spec.metric is None:
                continue
            history = self.tracker[spec.metric]
            self.table.update(spec.title, history[-1] if history else None)
        self.table.next_row()


class TrainValStage(Stage):
    """Batteries-included train + validation stage."""

    def __init__(self):
        super().__init__()
        self.is_train = True

    # -------------------------------------------------------- overridables

    def train_dataset(self):
        return self._lookup_dataset('train')

    def val_dataset(self):
        return self._lookup_dataset('val')

    def _lookup_dataset(self, key: str):
        ds = self.pipeline.datasets.get(key)
        if ds is None:
            raise ValueError(
                f'No "{key}" dataset found in pipeline. Use register_dataset("{key}", ...) to register a dataset.'
            )
        return ds

    def optimizers(self):
        return self.pipeline.optimizers.values()

    def loss_metric_name(self):
        return 'loss'

    def train_metric_prefix(self):
        return 'train'

    def val_metric_prefix(self):
        return 'val'

    def gradient_clip(self):
        """Max gradient norm; 0 disables clipping."""
        return 0.0

    def step(self, batch) -> torch.Tensor:
        raise NotImplementedError()

    def train_step(self, batch):
        return self.step(batch)

    def val_step(self, batch):
        return self.step(batch)

    # ------------------------------------------------------------ hot path

    def zero_grad(self):
        for optimizer in self.optimizers():
            optimizer.zero_grad()

    def clip_gradients(self):
        max_norm = self.gradient_clip()
        for optimizer in self.optimizers():
            if isinstance(optimizer, FlatOptimizer):
                # fused device-side global-norm clip, no host sync;
                # norm_scale inside accounts for the rank-summed grads
                optimizer.clip_grad_norm_(max_norm)
            else:
                for group in optimizer.param_groups:
                    torch.nn.utils.clip_grad_norm_(group['params'], max_norm)

    def _grad_sync(self):
        """Flat replicas need an explicit all-reduce (DDP syncs inside
        backward)."""
        for model in self.pipeline.models.values():
            if hasattr(model, 'grad_sync'):
                model.grad_sync()

    def optimize(self, loss):
        loss.backward()
        self._grad_sync()
        if self.gradient_clip():
            self.clip_gradients()
        for optimizer in self.optimizers():
            optimizer.step()

    def train_batch(self, batch):
        """One full training step: zero_grad -> forward -> backward ->
        (grad sync) -> optimizer, plus the standard step metrics. The
        benchmark driver calls this directly."""
        t0 = time.perf_counter_ns()
        with roctx_range('train_batch'):
            self.zero_grad()
            loss = self.train_step(batch)
            self.optimize(loss)
        elapsed_ms = (time.perf_counter_ns() - t0) / 1e6

        self.track_reduce(self.loss_metric_name(), loss)
        self._count_batch('train')
        self.track_reduce('misc/step_time_ms', torch.tensor(elapsed_ms), prefixed=False)

    def _count_batch(self, split: str):
        one = torch.tensor(1)
        self.track_reduce(f'misc/total_{split}_batches', one, reduction=Reduction.SUM, prefixed=False)
        self.track_reduce(
            f'misc/worker_{split}_batches',
            one,
            reduction=Reduction.SUM,
            reduce_globally=False,
            prefixed=False,
        )

    # --------------------------------------------------------- epoch loops

    def run_epoch(self):
        self.train_epoch()
        self.val_epoch()

    def _advance_sampler_epoch(self, loader):
        """Tell a DataLoader's DistributedSampler (or an epoch-aware
        dataset) which epoch this is, so shuffles differ per epoch."""
        sampler = getattr(loader, 'sampler', None)
        if sampler is not None and hasattr(sampler, 'set_epoch'):
            sampler.set_epoch(self.current_epoch)
        elif hasattr(loader, 'set_epoch'):
            loader.set_epoch(self.current_epoch)

    def train_epoch(self):
        self.is_train = True
        self.metric_prefix = self.train_metric_prefix()

        loader = self.train_dataset()
        self._advance_sampler_epoch(loader)
        for batch in loader:
            self.train_batch(batch)

        for name, scheduler in self.pipeline.schedulers.items():
            self.track(f'misc/lr_{name}', scheduler.get_last_lr()[0], prefixed=False)
            scheduler.step()

    @torch.no_grad()
    def val_epoch(self):
        self.is_train = False
        self.metric_prefix = self.val_metric_prefix()

        for batch in self.val_dataset():
            loss = self.val_step(batch)
            self.track_reduce(self.loss_metric_name(), loss)
            self._count_batch('val')

    def table_columns(self):
        columns = super().table_columns()
        columns.insert(1, {'name': '[Train] Loss', 'metric': f'{self.train_metric_prefix()}/{self.loss_metric_name()}'})
        columns.insert(2, {'name': '[Val] Loss', 'metric': f'{self.val_metric_prefix()}/{self.loss_metric_name()}'})
        return columns
