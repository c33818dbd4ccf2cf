"""GPT-2 small for the synthetic-token benchmark config (BASELINE.json).

Self-contained GPT-2 (124M): learned positional embeddings, pre-LN
blocks, GELU MLP, tied LM head. Attention uses
torch.nn.functional.scaled_dot_product_attention, which lowers to AOTriton
flash kernels on ROCm.
"""

import math
from dataclasses import dataclass

import torch
from torch import nn
from torch.nn import functional as F

from ..ops.fused_attn import sdpa
from ..ops.fused_ln import LayerNorm
from ..ops.fused_loss import cross_entropy


@dataclass
class GPT2Config:
    vocab_size: int = 50257
    n_positions: int = 1024
    n_embd: int = 768
    n_layer: int = 12
    n_head: int = 12
    dropout: float = 0.0


class CausalSelfAttention(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        assert cfg.n_embd % cfg.n_head == 0
        self.n_head = cfg.n_head
        self.c_attn = nn.Linear(cfg.n_embd, 3 * cfg.n_embd)
        self.c_proj = nn.Linear(cfg.n_embd, cfg.n_embd)

    def forward(self, x):
        B, T, C = x.shape
        qkv = self.c_attn(x)
        q, k, v = qkv.split(C, dim=2)
        q = q.view(B, T, self.n_head, C // self.n_head).transpose(1, 2)
        k = k.view(B, T, self.n_head, C // self.n_head).transpose(1, 2)
        v = v.view(B, T, self.n_head, C // self.n_head).transpose(1, 2)
        # custom MFMA flash attention on bf16/D=64 (ops/fused_attn.py);
        # beats AOTriton fwd+bwd combined at this shape — falls back to
        # torch SDPA otherwise
        y = sdpa(q, k, v, causal=True)
        y = y.transpose(1, 2).reshape(B, T, C)
        return self.c_proj(y)


class Block(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.ln_1 = LayerNorm(cfg.n_embd)
        self.attn = CausalSelfAttention(cfg)
        self.ln_2 = LayerNorm(cfg.n_embd)
        self.mlp = nn.Sequential(
            nn.Linear(cfg.n_embd, 4 * cfg.n_embd),
            nn.GELU(approximate='tanh'),
            nn.Linear(4 * cfg.n_embd, cfg.n_embd),
        )

    def forward(self, x):
        x = x + self.attn(self.ln_1(x))
        x = x + self.mlp(self.ln_2(x))
        return x


class GPT2(nn.Module):
    def __init__(self, cfg: GPT2Config = None):
        super().__init__()
        cfg = cfg or GPT2Config()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.n_embd)
        self.wpe = nn.Embedding(cfg.n_positions, cfg.n_embd)
        self.blocks = nn.ModuleList(Block(cfg) for _ in range(cfg.n_layer))
        self.ln_f = LayerNorm(cfg.n_embd)
        self.lm_head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=False)
        self.lm_head.weight = self.wte.weight  # tied

        self.apply(self._init_weights)
        # scaled init for residual projections (GPT-2 paper)
        for name, p in self.named_parameters():
            if name.endswith('c_proj.weight') or name.endswith('mlp.2.weight'):
                nn.init.normal_(p, mean=0.0, std=0.02 / math.sqrt(2 * cfg.n_layer))

    @staticmethod
    def _init_weights(module):
        if isinstance(module, nn.Linear):
            nn.init.normal_(module.weight, mean=0.0, std=0.02)
            if module.bias is not None:
                nn.init.zeros_(module.bias)
        elif isinstance(module, nn.Embedding):
            nn.init.normal_(module.weight, mean=0.0, std=0.02)

    def forward(self, idx, targets=None):
        B, T = idx.shape
        pos = torch.arange(T, device=idx.device)
        x = self.wte(idx) + self.wpe(pos)
        for block in self.blocks:
            x = block(x)
        x = self.ln_f(x)
        logits = self.lm_head(x)
        if targets is None:
            return logits
        # fused online-softmax CE on gfx950 for bf16 logits
        loss = cross_entropy(logits.view(-1, logits.size(-1)), targets.reshape(-1))
        return logits, loss


def gpt2_small() -> GPT2:
    return GPT2(GPT2Config())


def gpt2_tiny() -> GPT2:
    """Small config for CPU tests."""
    return GPT2(GPT2Config(vocab_size=512, n_positions=64, n_embd=64, n_layer=2, n_head=2))
