"""High-level API example: TrainValStage MNIST training.

Demonstrates the batteries-included path (capability mirror of reference
examples/mnist.py): register datasets/model/optimizer in ``pre_stage``,
implement ``step``, and let TrainValStage own the epoch loop, metric
reduction, progress table and checkpointing. Data is synthetic because
this environment has no network access — swap ``SyntheticMnist`` for
``torchvision.datasets.MNIST`` where downloads work.

Run:  python examples/mnist.py
      torchrun --standalone --nproc-per-node 8 examples/mnist.py
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader
from torch.utils.data.distributed import DistributedSampler

from dmlcloud_amd import TrainingPipeline, TrainValStage
from dmlcloud_amd.models import SyntheticMnist, mnist_cnn
from dmlcloud_amd.parallel import init_process_group_auto


def _loader(dataset, batch_size, shuffle):
    return DataLoader(dataset, batch_size=batch_size, sampler=DistributedSampler(dataset, shuffle=shuffle))


class MnistStage(TrainValStage):
    def __init__(self, batch_size=32):
        super().__init__()
        self.batch_size = batch_size

    def pre_stage(self):
        self.pipeline.register_dataset(
            'train', _loader(SyntheticMnist(n=8192), self.batch_size, shuffle=True)
        )
        self.pipeline.register_dataset(
            'val', _loader(SyntheticMnist(n=1024, seed=1), self.batch_size, shuffle=False)
        )
        # register_model moves to the device and wraps for data parallelism
        # (torch DDP here; pass ddp_impl='flat' for the flat-replica path)
        self.pipeline.register_model('cnn', mnist_cnn())
        self.pipeline.register_optimizer(
            'adam', torch.optim.Adam(self.pipeline.models['cnn'].parameters(), lr=1e-3)
        )

    def step(self, batch) -> torch.Tensor:
        img, target = (t.to(self.device) for t in batch)
        logits = self.pipeline.models['cnn'](img)
        hits = (logits.argmax(dim=1) == target).float().mean()
        self.track_reduce('accuracy', hits)
        return F.cross_entropy(logits, target)

    def table_columns(self):
        columns = super().table_columns()
        columns.insert(-2, {'name': '[Val] Acc.', 'metric': 'val/accuracy'})
        columns.insert(-2, {'name': '[Train] Acc.', 'metric': 'train/accuracy'})
        return columns


def main():
    cli = argparse.ArgumentParser(description=__doc__)
    cli.add_argument('--epochs', type=int, default=3)
    cli.add_argument('--batch-size', type=int, default=32)
    cli.add_argument('--checkpoints', default='checkpoints')
    args = cli.parse_args()

    init_process_group_auto()
    pipeline = TrainingPipeline(name='mnist')
    pipeline.enable_checkpointing(args.checkpoints, resume=False)
    pipeline.append_stage(MnistStage(args.batch_size), max_epochs=args.epochs)
    pipeline.run()


if __name__ == '__main__':
    main()
