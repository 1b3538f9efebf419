"""BERT-large encoder for the heterogeneous batch (BASELINE config 4).

Pre-LN encoder with bidirectional fused attention and masked-LM loss via
the fused cross-entropy kernel (shift=False; unmasked positions carry
ignore_index).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from saturn_amd.ops.functional import (
    FusedLayerNorm,
    full_attention,
    fused_cross_entropy,
    FusedEmbedding,
)


@dataclass
class BertConfig:
    vocab_size: int = 30522
    n_ctx: int = 512
    n_embd: int = 1024
    n_head: int = 16
    n_layer: int = 24


class BertBlock(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.ln_1 = FusedLayerNorm(cfg.n_embd)
        self.qkv = nn.Linear(cfg.n_embd, 3 * cfg.n_embd)
        self.proj = nn.Linear(cfg.n_embd, cfg.n_embd)
        self.ln_2 = FusedLayerNorm(cfg.n_embd)
        self.mlp = nn.Sequential(
            nn.Linear(cfg.n_embd, 4 * cfg.n_embd),
            nn.GELU(approximate="tanh"),
            nn.Linear(4 * cfg.n_embd, cfg.n_embd),
        )
        self.n_head = cfg.n_head
        self.head_dim = cfg.n_embd // cfg.n_head

    def forward(self, x):
        B, T, E = x.shape
        h = self.ln_1(x)
        q, k, v = self.qkv(h).split(E, dim=-1)
        q = q.view(B, T, self.n_head, self.head_dim).transpose(1, 2)
        k = k.view(B, T, self.n_head, self.head_dim).transpose(1, 2)
        v = v.view(B, T, self.n_head, self.head_dim).transpose(1, 2)
        o = full_attention(q, k, v).transpose(1, 2).reshape(B, T, E)
        x = x + self.proj(o)
        x = x + self.mlp(self.ln_2(x))
        return x


class BertForMaskedLM(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.wte = FusedEmbedding(cfg.vocab_size, cfg.n_embd)
        self.wpe = FusedEmbedding(cfg.n_ctx, cfg.n_embd)
        self.h = nn.ModuleList(BertBlock(cfg) for _ in range(cfg.n_layer))
        self.ln_f = FusedLayerNorm(cfg.n_embd)
        self.lm_head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=False)
        self.apply(self._init)

    @staticmethod
    def _init(m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, input_ids):
        B, T = input_ids.shape
        pos = torch.arange(T, device=input_ids.device)
        x = self.wte(input_ids) + self.wpe(pos)[None]
        for block in self.h:
            x = block(x)
        return self.lm_head(self.ln_f(x))


def mlm_loss(logits, labels):
    """labels: [B, T] with -100 at unmasked positions."""
    return fused_cross_entropy(logits, labels, shift=False, ignore_index=-100)


def get_bert_model(kwargs=None):
    kwargs = kwargs or {}
    cfg = BertConfig(
        n_layer=kwargs.get("n_layer", 24),
        n_ctx=kwargs.get("n_ctx", 512),
        vocab_size=kwargs.get("vocab_size", 30522),
        n_embd=kwargs.get("n_embd", 1024),
        n_head=kwargs.get("n_head", 16),
    )
    torch.manual_seed(kwargs.get("seed", 0))
    return BertForMaskedLM(cfg)


class SyntheticMLM(torch.utils.data.Dataset):
    """(masked_tokens, labels) pairs: 15% positions masked."""

    def __init__(self, n, seq_len, vocab, seed=1, mask_id=103):
        g = torch.Generator().manual_seed(seed)
        self.x = torch.randint(0, vocab, (n, seq_len), generator=g)
        self.mask = torch.rand(n, seq_len, generator=g) < 0.15
        self.mask_id = mask_id

    def __len__(self):
        return len(self.x)

    def __getitem__(self, i):
        x = self.x[i].clone()
        labels = torch.full_like(x, -100)
        labels[self.mask[i]] = x[self.mask[i]]
        x[self.mask[i]] = self.mask_id
        return x, labels


def make_mlm_dataloader(batch_size=16, seq_len=512, vocab=30522, n_batches=32):
    def get_dataloader():
        return torch.utils.data.DataLoader(
            SyntheticMLM(batch_size * n_batches, seq_len, vocab),
            batch_size=batch_size,
            shuffle=False,
        )

    return get_dataloader
