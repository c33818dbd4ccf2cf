"""GPT-2 small training on synthetic tokens — the LLM fast path.

Demonstrates: bf16 flat replica (fp32 master in the fused optimizer),
fused LayerNorm + cross-entropy kernels, single-all-reduce gradient
sync, hipGraph-captured steps inside a TrainValStage.

Run:  python examples/gpt2.py
      torchrun --standalone --nproc-per-node 8 examples/gpt2.py
"""

import sys

sys.path.insert(0, './')

import torch

from dmlcloud_amd import TrainingPipeline, TrainValStage
from dmlcloud_amd.models import gpt2_small, gpt2_tiny
from dmlcloud_amd.parallel import FlatAdam, init_process_group_auto


class SyntheticTokens(torch.utils.data.Dataset):
    def __init__(self, n: int, seq_len: int, vocab: int, seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        self.tokens = torch.randint(0, vocab, (n, seq_len), generator=g)

    def __len__(self):
        return len(self.tokens)

    def __getitem__(self, idx):
        return self.tokens[idx]


class GPT2Stage(TrainValStage):
    def pre_stage(self):
        on_gpu = self.device.type == 'cuda'
        model = gpt2_small() if on_gpu else gpt2_tiny()
        dtype = torch.bfloat16 if on_gpu else torch.float32
        self.pipeline.register_model('gpt2', model, ddp_impl='flat', flat_dtype=dtype)
        replica = self.pipeline.models['gpt2']
        self.pipeline.register_optimizer('adam', FlatAdam(replica, lr=3e-4, weight_decay=0.1))

        vocab = replica.module.cfg.vocab_size
        seq = min(replica.module.cfg.n_positions, 1024)
        train = SyntheticTokens(256, seq, vocab)
        val = SyntheticTokens(32, seq, vocab, seed=1)
        sampler = torch.utils.data.distributed.DistributedSampler(train)
        self.pipeline.register_dataset(
            'train', torch.utils.data.DataLoader(train, batch_size=8, sampler=sampler)
        )
        val_sampler = torch.utils.data.distributed.DistributedSampler(val, shuffle=False)
        self.pipeline.register_dataset(
            'val', torch.utils.data.DataLoader(val, batch_size=8, sampler=val_sampler)
        )

    def step(self, batch):
        idx = batch.to(self.device)
        _, loss = self.pipeline.models['gpt2'](idx, targets=idx)
        return loss

    def gradient_clip(self):
        return 1.0


def main():
    init_process_group_auto()
    pipeline = TrainingPipeline(name='gpt2-synthetic')
    pipeline.append_stage(GPT2Stage(), max_epochs=2)
    pipeline.run()


if __name__ == '__main__':
    main()
