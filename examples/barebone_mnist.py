"""Low-level API example: manual epoch loop on the base Stage class.

The capability mirror of reference examples/barebone_mnist.py, but on
the MI355X fast path: the model is a ``FlatReplica`` (all parameters in
one flat HBM buffer), gradients sync with a single RCCL all-reduce, and
the optimizer is the fused single-kernel ``FlatAdam``. Only ``run_epoch``
is implemented by hand — everything TrainValStage would otherwise do
(loops, loss tracking, prefixing) is spelled out here.

Run:  python examples/barebone_mnist.py
      torchrun --standalone --nproc-per-node 8 examples/barebone_mnist.py
"""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader
from torch.utils.data.distributed import DistributedSampler

from dmlcloud_amd import Stage, TrainingPipeline
from dmlcloud_amd.models import SyntheticMnist, mnist_cnn
from dmlcloud_amd.parallel import FlatAdam, FlatReplica, init_process_group_auto


class BareboneMnistStage(Stage):
    def pre_stage(self):
        train_set = SyntheticMnist(n=8192)
        self.train_sampler = DistributedSampler(train_set)
        self.train_loader = DataLoader(train_set, batch_size=32, sampler=self.train_sampler)

        val_set = SyntheticMnist(n=1024, seed=1)
        self.val_loader = DataLoader(val_set, batch_size=32, sampler=DistributedSampler(val_set, shuffle=False))

        # one flat fp32 parameter buffer; broadcast from rank 0 at init
        self.replica = FlatReplica(mnist_cnn().to(self.device))
        self.optimizer = FlatAdam(self.replica, lr=1e-3)

    def run_epoch(self):
        self.train_sampler.set_epoch(self.current_epoch)

        self.metric_prefix = 'train'
        self.replica.module.train()
        for batch in self.train_loader:
            self._observe(*self._forward(batch), train=True)

        self.metric_prefix = 'val'
        self.replica.module.eval()
        with torch.no_grad():
            for batch in self.val_loader:
                self._observe(*self._forward(batch), train=False)

    def _forward(self, batch):
        img, target = (t.to(self.device) for t in batch)
        logits = self.replica(img)
        return logits, target

    def _observe(self, logits, target, train):
        loss = F.cross_entropy(logits, target)
        if train:
            self.replica.zero_grad()
            loss.backward()
            self.replica.grad_sync()  # ONE RCCL all-reduce of the flat grads
            self.optimizer.step()  # fused gfx950 Adam kernel
        self.track_reduce('loss', loss)
        self.track_reduce('accuracy', (logits.argmax(dim=1) == target).float().mean())

    def table_columns(self):
        extra = [
            ('[Train] Loss', 'train/loss'),
            ('[Val] Loss', 'val/loss'),
            ('[Train] Acc.', 'train/accuracy'),
            ('[Val] Acc.', 'val/accuracy'),
        ]
        columns = super().table_columns()
        for offset, (title, metric) in enumerate(extra):
            columns.insert(1 + offset, {'name': title, 'metric': metric})
        return columns


def main():
    init_process_group_auto()
    pipeline = TrainingPipeline()
    pipeline.append_stage(BareboneMnistStage(), max_epochs=3)
    pipeline.run()


if __name__ == '__main__':
    main()
