"""Llama-3 family (8B / 70B) on saturn_amd ops.

BASELINE.json configs 3-5 workloads.  GQA attention (flash kernel,
head_dim 128), RMSNorm, SwiGLU MLP, half-split RoPE (theta 500000).
Random-init weights, synthetic tokens (no network for checkpoints).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from saturn_amd.ops.functional import (
    FusedRMSNorm,
    apply_rope,
    causal_attention,
    fused_cross_entropy,
    fused_swiglu,
    rope_tables,
    FusedEmbedding,
)


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    n_ctx: int = 8192
    n_embd: int = 4096
    n_head: int = 32
    n_kv_head: int = 8
    n_layer: int = 32
    ffn_dim: int = 14336
    rope_theta: float = 500000.0


PRESETS = {
    "8b": LlamaConfig(),
    "70b": LlamaConfig(
        n_embd=8192, n_head=64, n_kv_head=8, n_layer=80, ffn_dim=28672
    ),
    "1b-proxy": LlamaConfig(
        n_embd=2048, n_head=32, n_kv_head=8, n_layer=16, ffn_dim=8192
    ),
}


class LlamaAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.n_head = cfg.n_head
        self.n_kv = cfg.n_kv_head
        self.head_dim = cfg.n_embd // cfg.n_head
        E = cfg.n_embd
        self.q_proj = nn.Linear(E, self.n_head * self.head_dim, bias=False)
        self.k_proj = nn.Linear(E, self.n_kv * self.head_dim, bias=False)
        self.v_proj = nn.Linear(E, self.n_kv * self.head_dim, bias=False)
        self.o_proj = nn.Linear(self.n_head * self.head_dim, E, bias=False)
        # NOT buffers: model.to(bf16) would quantize the angle tables (see
        # gptj.py); fp32 [T, half] rows read directly by the fused kernel
        self._rope_f32 = rope_tables(cfg.n_ctx, self.head_dim,
                                     base=cfg.rope_theta)

    def _rope(self, device):
        if self._rope_f32[0].device != device:
            self._rope_f32 = tuple(t.to(device) for t in self._rope_f32)
        return self._rope_f32

    def forward(self, x):
        B, T, E = x.shape
        D = self.head_dim
        q = self.q_proj(x).view(B, T, self.n_head, D)
        k = self.k_proj(x).view(B, T, self.n_kv, D)
        v = self.v_proj(x).view(B, T, self.n_kv, D)
        cos, sin = self._rope(x.device)
        q = apply_rope(q, cos, sin, half_style=True)
        k = apply_rope(k, cos, sin, half_style=True)
        o = causal_attention(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2)
        )
        return self.o_proj(o.transpose(1, 2).reshape(B, T, -1))


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.gate_proj = nn.Linear(cfg.n_embd, cfg.ffn_dim, bias=False)
        self.up_proj = nn.Linear(cfg.n_embd, cfg.ffn_dim, bias=False)
        self.down_proj = nn.Linear(cfg.ffn_dim, cfg.n_embd, bias=False)

    def forward(self, x):
        return self.down_proj(fused_swiglu(self.gate_proj(x), self.up_proj(x)))


class LlamaBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.input_layernorm = FusedRMSNorm(cfg.n_embd)
        self.self_attn = LlamaAttention(cfg)
        self.post_attention_layernorm = FusedRMSNorm(cfg.n_embd)
        self.mlp = LlamaMLP(cfg)

    def forward(self, x):
        x = x + self.self_attn(self.input_layernorm(x))
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.wte = FusedEmbedding(cfg.vocab_size, cfg.n_embd)
        self.h = nn.ModuleList(LlamaBlock(cfg) for _ in range(cfg.n_layer))
        self.ln_f = FusedRMSNorm(cfg.n_embd)
        self.lm_head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=False)
        self.apply(self._init)

    @staticmethod
    def _init(m):
        if isinstance(m, nn.Linear):
            nn.init.normal_(m.weight, std=0.02)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=0.02)

    def forward(self, input_ids):
        x = self.wte(input_ids)
        for block in self.h:
            x = block(x)
        return self.lm_head(self.ln_f(x))


class _LlamaEmbed(nn.Module):
    def __init__(self, wte):
        super().__init__()
        self.wte = wte

    def forward(self, input_ids):
        return self.wte(input_ids)


class _LlamaHead(nn.Module):
    def __init__(self, ln_f, lm_head):
        super().__init__()
        self.ln_f = ln_f
        self.lm_head = lm_head

    def forward(self, x):
        return self.lm_head(self.ln_f(x))


def as_sequential(model: "LlamaForCausalLM") -> nn.Sequential:
    """Flatten for the pipeline executor (shared parameters)."""
    return nn.Sequential(
        _LlamaEmbed(model.wte), *model.h, _LlamaHead(model.ln_f, model.lm_head)
    )


def llama_loss(logits, labels):
    return fused_cross_entropy(logits, labels, shift=True)


def get_llama_model(kwargs=None):
    kwargs = kwargs or {}
    from dataclasses import replace

    cfg = PRESETS[kwargs.get("preset", "8b")]
    for key in ("n_layer", "n_ctx", "vocab_size"):
        if key in kwargs:
            cfg = replace(cfg, **{key: kwargs[key]})
    torch.manual_seed(kwargs.get("seed", 0))
    return LlamaForCausalLM(cfg)
