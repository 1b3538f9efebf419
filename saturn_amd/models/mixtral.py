"""Mixtral-style sparse MoE transformer on saturn_amd ops.

Beyond the reference's scope (SURVEY §2.2 lists EP as absent upstream):
Llama-architecture blocks whose FFN is a top-k routed mixture of SwiGLU
experts.  Pairs with the expert-parallel executor
(``saturn_amd.executors.expert`` / ``saturn_amd.parallel.expert``) which
shards experts across ranks with all-to-all token dispatch over RCCL —
this module alone is the single-process reference semantics.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from saturn_amd.models.llama import LlamaAttention
from saturn_amd.ops.functional import (
    FusedEmbedding,
    FusedRMSNorm,
    fused_cross_entropy,
    fused_swiglu,
)


@dataclass
class MixtralConfig:
    vocab_size: int = 32000
    n_ctx: int = 4096
    n_embd: int = 4096
    n_head: int = 32
    n_kv_head: int = 8
    n_layer: int = 32
    ffn_dim: int = 14336
    rope_theta: float = 1000000.0
    n_expert: int = 8
    top_k: int = 2


PRESETS = {
    "8x7b": MixtralConfig(),
    "8x7b-proxy": MixtralConfig(n_embd=2048, n_head=16, n_kv_head=8,
                                n_layer=8, ffn_dim=4096),
}


class MoEExpert(nn.Module):
    """One SwiGLU FFN expert (same shape as the Llama MLP)."""

    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.gate_proj = nn.Linear(cfg.n_embd, cfg.ffn_dim, bias=False)
        self.up_proj = nn.Linear(cfg.n_embd, cfg.ffn_dim, bias=False)
        self.down_proj = nn.Linear(cfg.ffn_dim, cfg.n_embd, bias=False)

    def forward(self, x):
        return self.down_proj(fused_swiglu(self.gate_proj(x), self.up_proj(x)))


def route(router: nn.Linear, x_flat: torch.Tensor, top_k: int):
    """Top-k routing: returns (weights [N,k] in x.dtype, expert ids [N,k]).

    Softmax in fp32 over all experts, then renormalized over the chosen k
    (Mixtral semantics).
    """
    logits = router(x_flat).float()
    probs = torch.softmax(logits, dim=-1)
    topv, topi = probs.topk(top_k, dim=-1)
    topv = topv / topv.sum(dim=-1, keepdim=True)
    return topv.to(x_flat.dtype), topi


class MoEMLP(nn.Module):
    """Dense-dispatch reference MoE layer (single process, exact)."""

    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.n_expert = cfg.n_expert
        self.top_k = cfg.top_k
        self.router = nn.Linear(cfg.n_embd, cfg.n_expert, bias=False)
        self.experts = nn.ModuleList(MoEExpert(cfg) for _ in range(cfg.n_expert))

    def forward(self, x):
        B, T, E = x.shape
        xf = x.reshape(-1, E)
        weights, topi = route(self.router, xf, self.top_k)
        out = torch.zeros_like(xf)
        for e, expert in enumerate(self.experts):
            tok, slot = (topi == e).nonzero(as_tuple=True)
            if tok.numel() == 0:
                continue
            out = out.index_add(
                0, tok, expert(xf[tok]) * weights[tok, slot].unsqueeze(-1)
            )
        return out.reshape(B, T, E)


class MixtralBlock(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.input_layernorm = FusedRMSNorm(cfg.n_embd)
        self.self_attn = LlamaAttention(cfg)
        self.post_attention_layernorm = FusedRMSNorm(cfg.n_embd)
        self.mlp = MoEMLP(cfg)

    def forward(self, x):
        x = x + self.self_attn(self.input_layernorm(x))
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class MixtralForCausalLM(nn.Module):
    def __init__(self, cfg: MixtralConfig):
        super().__init__()
        self.cfg = cfg
        self.wte = FusedEmbedding(cfg.vocab_size, cfg.n_embd)
        self.h = nn.ModuleList(MixtralBlock(cfg) for _ in range(cfg.n_layer))
        self.ln_f = FusedRMSNorm(cfg.n_embd)
        self.lm_head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=False)
        self.apply(self._init)

    @staticmethod
    def _init(m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)

    def forward(self, input_ids):
        x = self.wte(input_ids)
        for block in self.h:
            x = block(x)
        return self.lm_head(self.ln_f(x))

    def expert_parameters(self):
        """Parameters owned by experts (EP shards these; everything else is
        replicated and DDP-synced)."""
        out = []
        for block in self.h:
            for expert in block.mlp.experts:
                out.extend(expert.parameters())
        return out


def mixtral_loss(logits, labels):
    return fused_cross_entropy(logits, labels, shift=True)


def get_mixtral_model(kwargs=None):
    kwargs = kwargs or {}
    if "preset" in kwargs:
        cfg = PRESETS[kwargs["preset"]]
    else:
        cfg = MixtralConfig(
            vocab_size=kwargs.get("vocab_size", 32000),
            n_ctx=kwargs.get("n_ctx", 4096),
            n_embd=kwargs.get("n_embd", 4096),
            n_head=kwargs.get("n_head", 32),
            n_kv_head=kwargs.get("n_kv_head", 8),
            n_layer=kwargs.get("n_layer", 32),
            ffn_dim=kwargs.get("ffn_dim", 14336),
            n_expert=kwargs.get("n_expert", 8),
            top_k=kwargs.get("top_k", 2),
        )
    torch.manual_seed(kwargs.get("seed", 0))
    return MixtralForCausalLM(cfg)
