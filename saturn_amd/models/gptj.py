"""GPT-J-6B (the reference's benchmark workload) built on saturn_amd ops.

Capability parity with ``examples/wikitext103/models/GPTJ.py:25-526``
(hidden 4096, 16 heads, head_dim 256, 28 layers, rotary_dim 64 interleaved,
vocab 50400, ctx 512, parallel attention+MLP block), re-implemented:

- the forward is CORRECT (the reference discards every block's output,
  GPTJ.py:383-386 — SURVEY quirk #3);
- LayerNorm, RoPE, attention, loss are this framework's fused CDNA4 kernels
  (falling back to reference math on CPU);
- a ``flatten()`` -> nn.Sequential view is provided for the pipeline
  executor (the reference flattens for torchgpipe, GPTJ.py:511-521).

Synthetic-data factories at the bottom feed the benchmarks (no network for
datasets; BASELINE.md mandates synthetic + random init).
"""

from __future__ import annotations


from dataclasses import dataclass

import torch
import torch.nn as nn

from saturn_amd.ops.functional import (
    fused_gelu,
    FusedLayerNorm,
    apply_rope,
    causal_attention,
    fused_add3,
    fused_cross_entropy,
    rope_tables,
    FusedDropout,
    FusedEmbedding,
)


@dataclass
class GPTJConfig:
    vocab_size: int = 50400
    n_ctx: int = 512
    n_embd: int = 4096
    n_head: int = 16
    n_layer: int = 28
    rotary_dim: int = 64
    embd_pdrop: float = 0.0
    resid_pdrop: float = 0.0


class GPTJAttention(nn.Module):
    def __init__(self, cfg: GPTJConfig):
        super().__init__()
        self.n_head = cfg.n_head
        self.head_dim = cfg.n_embd // cfg.n_head
        self.rotary_dim = cfg.rotary_dim
        E = cfg.n_embd
        self.q_proj = nn.Linear(E, E, bias=False)
        self.k_proj = nn.Linear(E, E, bias=False)
        self.v_proj = nn.Linear(E, E, bias=False)
        self.out_proj = nn.Linear(E, E, bias=False)
        self.n_ctx = cfg.n_ctx
        # NOT buffers: model.to(bf16) would quantize the angle tables, and
        # the fused kernel wants the same fp32 [T, half] rows every call
        # (no per-call dtype convert / expand); device placement is lazy.
        self._rope_f32 = rope_tables(cfg.n_ctx, cfg.rotary_dim)

    def _rope(self, device):
        if self._rope_f32[0].device != device:
            self._rope_f32 = tuple(t.to(device) for t in self._rope_f32)
        return self._rope_f32

    def forward(self, x):
        B, T, E = x.shape
        H, D = self.n_head, self.head_dim
        q = self.q_proj(x).view(B, T, H, D)
        k = self.k_proj(x).view(B, T, H, D)
        v = self.v_proj(x).view(B, T, H, D)
        cos, sin = self._rope(x.device)  # fp32 [n_ctx, half]
        # partial rotary: the kernel rotates the first rotary_dim dims and
        # copies the rest through — no split+cat (the reference
        # materializes both halves and concatenates, GPTJ.py:255-259)
        q = apply_rope(q, cos, sin)
        k = apply_rope(k, cos, sin)
        q = q.transpose(1, 2)  # [B, H, T, D]
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        o = causal_attention(q, k, v)
        o = o.transpose(1, 2).reshape(B, T, -1)  # -1: heads may be TP-sharded
        return self.out_proj(o)


class GPTJMLP(nn.Module):
    def __init__(self, cfg: GPTJConfig):
        super().__init__()
        E = cfg.n_embd
        self.fc_in = nn.Linear(E, 4 * E)
        self.fc_out = nn.Linear(4 * E, E)

    def forward(self, x):
        return self.fc_out(fused_gelu(self.fc_in(x)))


class GPTJBlock(nn.Module):
    """Parallel attention + MLP off one LayerNorm (GPT-J architecture)."""

    def __init__(self, cfg: GPTJConfig):
        super().__init__()
        self.ln_1 = FusedLayerNorm(cfg.n_embd)
        self.attn = GPTJAttention(cfg)
        self.mlp = GPTJMLP(cfg)
        # reference resid_pdrop (GPTJ.py:95-96); 0.0 = exact passthrough
        self.drop = FusedDropout(cfg.resid_pdrop)

    def forward(self, x):
        h = self.ln_1(x)
        return fused_add3(x, self.drop(self.attn(h)), self.drop(self.mlp(h)))


class GPTJForCausalLM(nn.Module):
    def __init__(self, cfg: GPTJConfig):
        super().__init__()
        self.cfg = cfg
        self.wte = FusedEmbedding(cfg.vocab_size, cfg.n_embd)
        self.drop = FusedDropout(cfg.embd_pdrop)
        self.h = nn.ModuleList(GPTJBlock(cfg) for _ in range(cfg.n_layer))
        self.ln_f = FusedLayerNorm(cfg.n_embd)
        self.lm_head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=True)
        self.apply(self._init)

    @staticmethod
    def _init(m):
        if isinstance(m, nn.Linear):
            nn.init.normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=0.02)

    def forward(self, input_ids):
        x = self.drop(self.wte(input_ids))
        for block in self.h:
            x = block(x)
        x = self.ln_f(x)
        return self.lm_head(x)


def pretraining_loss(logits, labels):
    """Shifted causal-LM loss via the fused CE kernel (reference
    GPTJ.py:491-499)."""
    return fused_cross_entropy(logits, labels, shift=True)


class _EmbedStage(nn.Module):
    def __init__(self, wte):
        super().__init__()
        self.wte = wte

    def forward(self, input_ids):
        return self.wte(input_ids)


class _HeadStage(nn.Module):
    def __init__(self, ln_f, lm_head):
        super().__init__()
        self.ln_f = ln_f
        self.lm_head = lm_head

    def forward(self, x):
        return self.lm_head(self.ln_f(x))


def as_sequential(model: GPTJForCausalLM) -> nn.Sequential:
    """Flatten into an nn.Sequential for the pipeline executor (parameters
    are shared with ``model``; the reference flattens GPT-J the same way for
    torchgpipe, GPTJ.py:511-521)."""
    return nn.Sequential(
        _EmbedStage(model.wte), *model.h, _HeadStage(model.ln_f, model.lm_head)
    )


# ---------------------------------------------------------------------------
# Factories (Task contract: picklable top-level callables)
# ---------------------------------------------------------------------------
def get_gptj_model(kwargs=None):
    kwargs = kwargs or {}
    cfg = GPTJConfig(
        n_layer=kwargs.get("n_layer", 28),
        n_embd=kwargs.get("n_embd", 4096),
        n_head=kwargs.get("n_head", 16),
        n_ctx=kwargs.get("n_ctx", 512),
        vocab_size=kwargs.get("vocab_size", 50400),
        rotary_dim=kwargs.get("rotary_dim", 64),
        embd_pdrop=kwargs.get("embd_pdrop", 0.0),
        resid_pdrop=kwargs.get("resid_pdrop", 0.0),
    )
    torch.manual_seed(kwargs.get("seed", 0))
    return GPTJForCausalLM(cfg)


class SyntheticTokens(torch.utils.data.Dataset):
    """Random token sequences; collate mirrors the reference's
    (batch, batch.clone()) contract (dataloaders.py:22-24)."""

    def __init__(self, n: int, seq_len: int, vocab: int, seed: int = 1):
        g = torch.Generator().manual_seed(seed)
        self.data = torch.randint(0, vocab, (n, seq_len), generator=g)

    def __len__(self):
        return len(self.data)

    def __getitem__(self, i):
        return self.data[i]


def _collate(batch):
    x = torch.stack(batch)
    return x, x.clone()


def make_token_dataloader(batch_size=8, seq_len=512, vocab=50400, n_batches=64):
    def get_dataloader():
        return torch.utils.data.DataLoader(
            SyntheticTokens(batch_size * n_batches, seq_len, vocab),
            batch_size=batch_size,
            shuffle=False,
            collate_fn=_collate,
        )

    return get_dataloader
