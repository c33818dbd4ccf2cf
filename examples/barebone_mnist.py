"""Low-level API example: manual epoch loop on the base Stage class.

Mirror of the reference example (reference examples/barebone_mnist.py)
on synthetic data, using the flat-replica fast path and fused Adam.
"""

import sys

sys.path.insert(0, './')

import torch
from torch import nn
from torch.utils.data import DataLoader

from dmlcloud_amd import Stage, TrainingPipeline
from dmlcloud_amd.models import SyntheticMnist, mnist_cnn
from dmlcloud_amd.parallel import FlatAdam, FlatReplica, init_process_group_auto


class MNISTStage(Stage):
    def pre_stage(self):
        train_dataset = SyntheticMnist(n=8192)
        self.train_sampler = torch.utils.data.distributed.DistributedSampler(train_dataset)
        self.train_loader = DataLoader(train_dataset, batch_size=32, sampler=self.train_sampler)

        val_dataset = SyntheticMnist(n=1024, seed=1)
        val_sampler = torch.utils.data.distributed.DistributedSampler(val_dataset, shuffle=False)
        self.val_loader = DataLoader(val_dataset, batch_size=32, sampler=val_sampler)

        self.replica = FlatReplica(mnist_cnn().to(self.pipeline.device))
        self.optimizer = FlatAdam(self.replica, lr=1e-3)
        self.loss = nn.CrossEntropyLoss()

    def run_epoch(self):
        self._train_epoch()
        self._val_epoch()

    def _train_epoch(self):
        self.replica.module.train()
        self.metric_prefix = 'train'
        self.train_sampler.set_epoch(self.current_epoch)

        for img, target in self.train_loader:
            img, target = img.to(self.pipeline.device), target.to(self.pipeline.device)
            self.replica.zero_grad()
            output = self.replica(img)
            loss = self.loss(output, target)
            loss.backward()
            self.replica.grad_sync()
            self.optimizer.step()
            self._log_metrics(output, target, loss)

    @torch.no_grad()
    def _val_epoch(self):
        self.replica.module.eval()
        self.metric_prefix = 'val'
        for img, target in self.val_loader:
            img, target = img.to(self.pipeline.device), target.to(self.pipeline.device)
            output = self.replica(img)
            loss = self.loss(output, target)
            self._log_metrics(output, target, loss)

    def _log_metrics(self, output, target, loss):
        self.track_reduce('loss', loss)
        self.track_reduce('accuracy', (output.argmax(1) == target).float().mean())

    def table_columns(self):
        columns = super().table_columns()
        columns.insert(1, {'name': '[Train] Loss', 'metric': 'train/loss'})
        columns.insert(2, {'name': '[Val] Loss', 'metric': 'val/loss'})
        columns.insert(3, {'name': '[Train] Acc.', 'metric': 'train/accuracy'})
        columns.insert(4, {'name': '[Val] Acc.', 'metric': 'val/accuracy'})
        return columns


def main():
    init_process_group_auto()
    pipeline = TrainingPipeline()
    pipeline.append_stage(MNISTStage(), max_epochs=3)
    pipeline.run()


if __name__ == '__main__':
    main()
