"""GPT-2 family (small / XL) on saturn_amd ops.

BASELINE.json config 2 workload ("4-job GPT-2-small HPO sweep") and part of
the heterogeneous batch (config 4: GPT-2-XL).  Standard pre-LN transformer:
learned position embeddings, serial attention->MLP blocks, weight-tied
LM head.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from saturn_amd.ops.functional import (
    FusedLayerNorm,
    causal_attention,
    fused_cross_entropy,
    FusedEmbedding,
)


@dataclass
class GPT2Config:
    vocab_size: int = 50257
    n_ctx: int = 1024
    n_embd: int = 768
    n_head: int = 12
    n_layer: int = 12


PRESETS = {
    "small": GPT2Config(),
    "medium": GPT2Config(n_embd=1024, n_head=16, n_layer=24),
    "large": GPT2Config(n_embd=1280, n_head=20, n_layer=36),
    "xl": GPT2Config(n_embd=1600, n_head=25, n_layer=48),
}


class GPT2Attention(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.n_head = cfg.n_head
        self.head_dim = cfg.n_embd // cfg.n_head
        self.c_attn = nn.Linear(cfg.n_embd, 3 * cfg.n_embd)
        self.c_proj = nn.Linear(cfg.n_embd, cfg.n_embd)

    def forward(self, x):
        B, T, E = x.shape
        H, D = self.n_head, self.head_dim
        q, k, v = self.c_attn(x).split(E, dim=-1)
        q = q.view(B, T, H, D).transpose(1, 2)
        k = k.view(B, T, H, D).transpose(1, 2)
        v = v.view(B, T, H, D).transpose(1, 2)
        o = causal_attention(q, k, v)
        return self.c_proj(o.transpose(1, 2).reshape(B, T, E))


class GPT2Block(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.ln_1 = FusedLayerNorm(cfg.n_embd)
        self.attn = GPT2Attention(cfg)
        self.ln_2 = FusedLayerNorm(cfg.n_embd)
        self.mlp = nn.Sequential(
            nn.Linear(cfg.n_embd, 4 * cfg.n_embd),
            nn.GELU(approximate="tanh"),
            nn.Linear(4 * cfg.n_embd, cfg.n_embd),
        )

    def forward(self, x):
        x = x + self.attn(self.ln_1(x))
        x = x + self.mlp(self.ln_2(x))
        return x


class GPT2ForCausalLM(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.cfg = cfg
        self.wte = FusedEmbedding(cfg.vocab_size, cfg.n_embd)
        self.wpe = FusedEmbedding(cfg.n_ctx, cfg.n_embd)
        self.h = nn.ModuleList(GPT2Block(cfg) for _ in range(cfg.n_layer))
        self.ln_f = FusedLayerNorm(cfg.n_embd)
        self.lm_head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=False)
        self.lm_head.weight = self.wte.weight  # weight tying
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
        B, T = input_ids.shape
        pos = torch.arange(T, device=input_ids.device)
        x = self.wte(input_ids) + self.wpe(pos)[None]
        for block in self.h:
            x = block(x)
        return self.lm_head(self.ln_f(x))


class _GPT2Embed(nn.Module):
    def __init__(self, wte, wpe):
        super().__init__()
        self.wte = wte
        self.wpe = wpe

    def forward(self, input_ids):
        pos = torch.arange(input_ids.shape[1], device=input_ids.device)
        return self.wte(input_ids) + self.wpe(pos)[None]


class _GPT2Head(nn.Module):
    def __init__(self, ln_f, lm_head):
        super().__init__()
        self.ln_f = ln_f
        self.lm_head = lm_head

    def forward(self, x):
        return self.lm_head(self.ln_f(x))


def as_sequential(model: "GPT2ForCausalLM") -> nn.Sequential:
    """Flatten for the pipeline executor (shared parameters)."""
    return nn.Sequential(
        _GPT2Embed(model.wte, model.wpe), *model.h,
        _GPT2Head(model.ln_f, model.lm_head)
    )


def gpt2_loss(logits, labels):
    return fused_cross_entropy(logits, labels, shift=True)


def get_gpt2_model(kwargs=None):
    kwargs = kwargs or {}
    preset = kwargs.get("preset", "small")
    cfg = PRESETS[preset]
    if "n_layer" in kwargs:
        from dataclasses import replace

        cfg = replace(cfg, n_layer=kwargs["n_layer"])
    if "n_ctx" in kwargs:
        from dataclasses import replace

        cfg = replace(cfg, n_ctx=kwargs["n_ctx"])
    torch.manual_seed(kwargs.get("seed", 0))
    return GPT2ForCausalLM(cfg)
