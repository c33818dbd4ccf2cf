"""ResNet-50 training on synthetic ImageNet-shaped data.

Demonstrates the bf16-autocast + channels-last + SyncBN + DDP path on
the self-contained ResNet-50 (dmlcloud_amd.models.resnet).

Run:  python examples/resnet50.py
      torchrun --standalone --nproc-per-node 8 examples/resnet50.py
"""

import sys

sys.path.insert(0, './')

import torch
from torch import nn
from torch.utils.data import DataLoader

from dmlcloud_amd import TrainingPipeline, TrainValStage
from dmlcloud_amd.models import resnet50
from dmlcloud_amd.parallel import init_process_group_auto


class SyntheticImageNet(torch.utils.data.Dataset):
    def __init__(self, n: int = 2048, seed: int = 0):
        self.n = n
        self.seed = seed

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed * 100003 + idx)
        return torch.randn(3, 224, 224, generator=g), int(torch.randint(0, 1000, (1,), generator=g))


class ResNetStage(TrainValStage):
    def pre_stage(self):
        on_gpu = self.device.type == 'cuda'
        train = SyntheticImageNet(512)
        val = SyntheticImageNet(128, seed=1)
        sampler = torch.utils.data.distributed.DistributedSampler(train)
        self.pipeline.register_dataset('train', DataLoader(train, batch_size=64, sampler=sampler, num_workers=2))
        val_sampler = torch.utils.data.distributed.DistributedSampler(val, shuffle=False)
        self.pipeline.register_dataset('val', DataLoader(val, batch_size=64, sampler=val_sampler))

        model = resnet50()
        if on_gpu:
            model = model.to(memory_format=torch.channels_last)
            torch.backends.cudnn.benchmark = True  # MIOpen find mode
        self.pipeline.register_model('resnet', model, sync_bn=on_gpu)
        self.pipeline.register_optimizer(
            'sgd', torch.optim.SGD(self.pipeline.models['resnet'].parameters(), lr=0.1, momentum=0.9)
        )
        self.loss = nn.CrossEntropyLoss()
        self.autocast = on_gpu

    def step(self, batch):
        img, target = batch
        img = img.to(self.device, non_blocking=True)
        if self.autocast:
            img = img.to(memory_format=torch.channels_last)
        target = target.to(self.device, non_blocking=True)
        with torch.autocast('cuda', dtype=torch.bfloat16, enabled=self.autocast):
            out = self.pipeline.models['resnet'](img)
            loss = self.loss(out, target)
        self.track_reduce('accuracy', (out.argmax(1) == target).float().mean())
        return loss

    def gradient_clip(self):
        return 5.0


def main():
    init_process_group_auto()
    pipeline = TrainingPipeline(name='resnet50-synthetic')
    pipeline.append_stage(ResNetStage(), max_epochs=2)
    pipeline.run()


if __name__ == '__main__':
    main()
