"""High-level API example: TrainValStage MNIST training.

Mirror of the reference example (reference examples/mnist.py) on
synthetic data (this environment has no network access for dataset
downloads; swap SyntheticMnist for torchvision.datasets.MNIST when
available).

Run:  python examples/mnist.py
      torchrun --standalone --nproc-per-node 8 examples/mnist.py
"""

import sys

sys.path.insert(0, './')

import torch
from torch import nn
from torch.utils.data import DataLoader

from dmlcloud_amd import TrainingPipeline, TrainValStage
from dmlcloud_amd.models import SyntheticMnist, mnist_cnn
from dmlcloud_amd.parallel import init_process_group_auto


class MNISTStage(TrainValStage):
    def pre_stage(self):
        train_dataset = SyntheticMnist(n=8192)
        train_sampler = torch.utils.data.distributed.DistributedSampler(train_dataset)
        self.pipeline.register_dataset('train', DataLoader(train_dataset, batch_size=32, sampler=train_sampler))

        val_dataset = SyntheticMnist(n=1024, seed=1)
        val_sampler = torch.utils.data.distributed.DistributedSampler(val_dataset, shuffle=False)
        self.pipeline.register_dataset('val', DataLoader(val_dataset, batch_size=32, sampler=val_sampler))

        self.pipeline.register_model('cnn', mnist_cnn())
        model = self.pipeline.models['cnn']
        self.pipeline.register_optimizer('adam', torch.optim.Adam(model.parameters(), lr=1e-3))
        self.loss = nn.CrossEntropyLoss()

    def step(self, batch) -> torch.Tensor:
        img, target = batch
        img, target = img.to(self.device), target.to(self.device)
        output = self.pipeline.models['cnn'](img)
        loss = self.loss(output, target)
        self.track_reduce('accuracy', (output.argmax(1) == target).float().mean())
        return loss

    def table_columns(self):
        columns = super().table_columns()
        columns.insert(-2, {'name': '[Val] Acc.', 'metric': 'val/accuracy'})
        columns.insert(-2, {'name': '[Train] Acc.', 'metric': 'train/accuracy'})
        return columns


def main():
    init_process_group_auto()
    pipeline = TrainingPipeline(name='mnist')
    pipeline.enable_checkpointing('checkpoints', resume=False)
    pipeline.append_stage(MNISTStage(), max_epochs=3)
    pipeline.run()


if __name__ == '__main__':
    main()
