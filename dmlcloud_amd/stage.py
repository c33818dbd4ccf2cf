"""Stage engine: epoch state machine with user hook points.

API parity with the reference Stage/TrainValStage (reference
dmlcloud/stage.py:18-341): same hook order (pre_stage -> [pre_epoch ->
run_epoch -> post_epoch]* -> post_stage), same metric namespacing
(train/ val/ misc/), same table semantics and stop_stage behavior.

Differences by design:
- progress table is the dependency-free utils/table.py implementation,
- per-step metric tracking feeds device-resident accumulators (no D2H
  per batch - see metrics.py),
- clip_gradients() computes a fused global-norm clip on flat optimizers
  (single gfx950 kernel chain, no host sync) and falls back to
  torch.nn.utils for stock optimizers.
"""

import sys
import time
from datetime import datetime
from typing import Any, Dict, List, Optional, Union

import torch

from .metrics import MetricTracker, Reduction
from .parallel.distributed import is_root
from .parallel.flat import FlatOptimizer
from .utils.logging import DevNullIO, flush_log_handlers
from .utils.table import ProgressTable
from .utils.tracing import roctx_range

__all__ = ['Stage', 'TrainValStage']


class Stage:
    """Hook points: pre_stage, post_stage, pre_epoch, post_epoch."""

    def __init__(self):
        self.pipeline = None  # set by the pipeline on append_stage
        self.max_epochs = None
        self.name = None

        self.start_time = None
        self.stop_time = None
        self.epoch_start_time = None
        self.epoch_stop_time = None
        self.current_epoch = 1
        self._stop_requested = False

        self.metric_prefix = None
        self.table = None
        self.barrier_timeout = None

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
        if prefixed and self.metric_prefix:
            name = f'{self.metric_prefix}/{name}'
        self.pipeline.track_reduce(name, value, step, reduction, dim, reduce_globally)

    def track(self, name: str, value, step: Optional[int] = None, prefixed: bool = True):
        if prefixed and self.metric_prefix:
            name = f'{self.metric_prefix}/{name}'
        self.pipeline.track(name, value, step)

    def stop_stage(self):
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

    def run(self):
        """Run until max_epochs or stop_stage()."""
        self._pre_stage()
        while self.max_epochs is None or self.current_epoch <= self.max_epochs:
            self._pre_epoch()
            with roctx_range(f'epoch_{self.current_epoch}'):
                self.run_epoch()
            self._post_epoch()
            if self._stop_requested:
                break
        self._post_stage()

    def _pre_stage(self):
        self.start_time = datetime.now()
        self.table = ProgressTable(file=sys.stdout if is_root() else DevNullIO())
        self._setup_table()
        if len(self.pipeline.stages) > 1:
            self.logger.info(f'\n========== STAGE: {self.name} ==========')
        self.pre_stage()
        flush_log_handlers(self.logger)
        self.pipeline.barrier(self.barrier_timeout)

    def _post_stage(self):
        self.table.close()
        self.post_stage()
        self.pipeline.barrier(self.barrier_timeout)
        self.stop_time = datetime.now()
        if len(self.pipeline.stages) > 1:
            self.logger.info(f'Finished stage in {self.stop_time - self.start_time}')

    def _pre_epoch(self):
        self.epoch_start_time = datetime.now()
        self.table['Epoch'] = self.current_epoch
        self.pre_epoch()
        self.pipeline._pre_epoch()

    def _post_epoch(self):
        self.epoch_stop_time = datetime.now()
        self._reduce_metrics()
        self.post_epoch()
        self.pipeline._post_epoch()
        self._update_table()
        self.current_epoch += 1

    def _reduce_metrics(self):
        self.track(name='misc/epoch', value=self.current_epoch, prefixed=False)
        self.track(
            name='misc/epoch_time',
            value=(self.epoch_stop_time - self.epoch_start_time).total_seconds(),
            prefixed=False,
        )
        self.tracker.next_epoch()

    def _setup_table(self):
        for column_dct in self._metrics():
            display_name = column_dct.pop('name')
            column_dct.pop('metric')
            self.table.add_column(display_name, **column_dct)

    def _update_table(self):
        self.table.update('Epoch', self.current_epoch)
        self.table.update('Time/Epoch', (datetime.now() - self.start_time) / self.current_epoch)
        if self.max_epochs is not None:
            self.table.update(
                'ETA',
                (datetime.now() - self.start_time) / self.current_epoch * (self.max_epochs - self.current_epoch),
            )
        for column_dct in self._metrics():
            display_name = column_dct['name']
            metric_name = column_dct['metric']
            if metric_name is not None:
                history = self.tracker[metric_name]
                self.table.update(display_name, history[-1] if history else None)
        self.table.next_row()

    def _metrics(self):
        metrics = []
        for column in self.table_columns():
            if isinstance(column, str):
                metrics.append({'name': column, 'metric': column})
            elif isinstance(column, dict):
                if 'name' not in column:
                    raise ValueError('Column dict must contain a "name" key')
                if 'metric' not in column:
                    raise ValueError('Column dict must contain a "metric" key')
                metrics.append(dict(column))
            else:
                raise ValueError(f'Invalid column: {column}. Must be a string or a dict.')
        return metrics


class TrainValStage(Stage):
    """Batteries-included train + validation stage."""

    def __init__(self):
        super().__init__()
        self.is_train = True

    def train_dataset(self):
        train_ds = self.pipeline.datasets.get('train')
        if train_ds is None:
            raise ValueError(
                'No "train" dataset found in pipeline. Use register_dataset("train", ...) to register a dataset.'
            )
        return train_ds

    def val_dataset(self):
        val_ds = self.pipeline.datasets.get('val')
        if val_ds is None:
            raise ValueError(
                'No "val" dataset found in pipeline. Use register_dataset("val", ...) to register a dataset.'
            )
        return val_ds

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

    def run_epoch(self):
        self.train_epoch()
        self.val_epoch()

    def step(self, batch) -> torch.Tensor:
        raise NotImplementedError()

    def train_step(self, batch):
        return self.step(batch)

    def val_step(self, batch):
        return self.step(batch)

    def zero_grad(self):
        for optimizer in self.optimizers():
            optimizer.zero_grad()

    def clip_gradients(self):
        max_norm = self.gradient_clip()
        for optimizer in self.optimizers():
            if isinstance(optimizer, FlatOptimizer):
                # fused device-side global-norm clip, no host sync
                optimizer.clip_grad_norm_(max_norm)
            else:
                for group in optimizer.param_groups:
                    torch.nn.utils.clip_grad_norm_(group['params'], max_norm)

    def optimize(self, loss):
        loss.backward()
        self._grad_sync()
        if self.gradient_clip():
            self.clip_gradients()
        for optimizer in self.optimizers():
            optimizer.step()

    def _grad_sync(self):
        """Flat replicas need an explicit all-reduce (DDP syncs inside
        backward)."""
        for model in self.pipeline.models.values():
            if hasattr(model, 'grad_sync'):
                model.grad_sync()

    def train_batch(self, batch):
        """One full training step: zero_grad -> forward -> backward ->
        (grad sync) -> optimizer, plus the standard step metrics. The
        benchmark driver calls this directly."""
        step_start_time = time.perf_counter_ns()
        with roctx_range('train_batch'):
            self.zero_grad()
            loss = self.train_step(batch)
            self.optimize(loss)
        step_end_time = time.perf_counter_ns()

        self.track_reduce(self.loss_metric_name(), loss)
        self.track_reduce('misc/total_train_batches', torch.tensor(1), reduction=Reduction.SUM, prefixed=False)
        self.track_reduce(
            'misc/worker_train_batches',
            torch.tensor(1),
            reduction=Reduction.SUM,
            reduce_globally=False,
            prefixed=False,
        )
        self.track_reduce('misc/step_time_ms', torch.tensor(step_end_time - step_start_time) / 1e6, prefixed=False)

    def train_epoch(self):
        self.is_train = True
        self.metric_prefix = self.train_metric_prefix()

        train_ds = self.train_dataset()
        if hasattr(train_ds, 'sampler') and hasattr(train_ds.sampler, 'set_epoch'):
            train_ds.sampler.set_epoch(self.current_epoch)
        elif hasattr(train_ds, 'set_epoch'):
            train_ds.set_epoch(self.current_epoch)

        for batch in train_ds:
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
            self.track_reduce('misc/total_val_batches', torch.tensor(1), reduction=Reduction.SUM, prefixed=False)
            self.track_reduce(
                'misc/worker_val_batches',
                torch.tensor(1),
                reduction=Reduction.SUM,
                reduce_globally=False,
                prefixed=False,
            )

    def table_columns(self):
        columns = super().table_columns()
        columns.insert(1, {'name': '[Train] Loss', 'metric': f'{self.train_metric_prefix()}/{self.loss_metric_name()}'})
        columns.insert(2, {'name': '[Val] Loss', 'metric': f'{self.val_metric_prefix()}/{self.loss_metric_name()}'})
        return columns
