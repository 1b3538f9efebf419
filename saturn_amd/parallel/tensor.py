"""Megatron-style tensor parallelism over RCCL/xGMI.

The reference only *declares* this technique (``Techniques.MEGATRON = 4``,
saturn/core/representations/Strategy.py:34) with no implementation; here it
is a real executor option: column/row-parallel linears with autograd-correct
collectives, head-sharded attention, and in-place sharding of the existing
model families (GPT-J / GPT-2 / Llama blocks).

Pure TP: every rank sees the same batch; row-parallel outputs are
all-reduced in forward, so replicated parameters (embeddings, norms)
receive identical gradients on every rank and need no extra sync.
"""

from __future__ import annotations



import torch
import torch.distributed as dist
import torch.nn as nn


def _world() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def _rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


class _CopyToTP(torch.autograd.Function):
    """Identity forward; all-reduce gradient (input of column-parallel)."""

    @staticmethod
    def forward(ctx, x):
        return x

    @staticmethod
    def backward(ctx, g):
        if _world() > 1:
            g = g.contiguous()
            dist.all_reduce(g)
        return g


class _ReduceFromTP(torch.autograd.Function):
    """All-reduce forward; identity gradient (output of row-parallel)."""

    @staticmethod
    def forward(ctx, x):
        if _world() > 1:
            x = x.contiguous()
            dist.all_reduce(x)
        return x

    @staticmethod
    def backward(ctx, g):
        return g


class ColumnParallelLinear(nn.Module):
    """y_local = x W_l^T + b_l; output dim sharded across ranks."""

    def __init__(self, linear: nn.Linear):
        super().__init__()
        w, r = _world(), _rank()
        out_f, in_f = linear.weight.shape
        assert out_f % w == 0, f"out_features {out_f} % tp {w} != 0"
        sl = slice(r * out_f // w, (r + 1) * out_f // w)
        self.weight = nn.Parameter(linear.weight.data[sl].clone())
        self.bias = (
            nn.Parameter(linear.bias.data[sl].clone())
            if linear.bias is not None
            else None
        )
        self.full_out = out_f

    def forward(self, x):
        return torch.nn.functional.linear(_CopyToTP.apply(x), self.weight, self.bias)


class RowParallelLinear(nn.Module):
    """y = sum_ranks x_l W_l^T (+ b on rank 0); input dim sharded."""

    def __init__(self, linear: nn.Linear):
        super().__init__()
        w, r = _world(), _rank()
        out_f, in_f = linear.weight.shape
        assert in_f % w == 0, f"in_features {in_f} % tp {w} != 0"
        sl = slice(r * in_f // w, (r + 1) * in_f // w)
        self.weight = nn.Parameter(linear.weight.data[:, sl].clone())
        # bias added once, post-reduce
        self.bias = (
            nn.Parameter(linear.bias.data.clone())
            if linear.bias is not None
            else None
        )

    def forward(self, x_local):
        y = torch.nn.functional.linear(x_local, self.weight)
        y = _ReduceFromTP.apply(y)
        if self.bias is not None:
            y = y + self.bias
        return y


def tp_shard_model(model: nn.Module) -> nn.Module:
    """In-place shard of the known block families.  Weights must already be
    rank-identical (broadcast before calling)."""
    from saturn_amd.models.gptj import GPTJBlock
    from saturn_amd.models.llama import LlamaBlock

    w = _world()
    if w == 1:
        return model
    for m in model.modules():
        if isinstance(m, GPTJBlock):
            a = m.attn
            assert a.n_head % w == 0
            a.q_proj = ColumnParallelLinear(a.q_proj)
            a.k_proj = ColumnParallelLinear(a.k_proj)
            a.v_proj = ColumnParallelLinear(a.v_proj)
            a.out_proj = RowParallelLinear(a.out_proj)
            a.n_head //= w
            m.mlp.fc_in = ColumnParallelLinear(m.mlp.fc_in)
            m.mlp.fc_out = RowParallelLinear(m.mlp.fc_out)
        elif isinstance(m, LlamaBlock):
            a = m.self_attn
            assert a.n_head % w == 0 and a.n_kv % w == 0
            a.q_proj = ColumnParallelLinear(a.q_proj)
            a.k_proj = ColumnParallelLinear(a.k_proj)
            a.v_proj = ColumnParallelLinear(a.v_proj)
            a.o_proj = RowParallelLinear(a.o_proj)
            a.n_head //= w
            a.n_kv //= w
            m.mlp.gate_proj = ColumnParallelLinear(m.mlp.gate_proj)
            m.mlp.up_proj = ColumnParallelLinear(m.mlp.up_proj)
            m.mlp.down_proj = RowParallelLinear(m.mlp.down_proj)
    return model


def tp_replicated_parameters(model: nn.Module):
    """Parameters that are replicated (not sharded) under TP: embeddings,
    norms, lm_head, and row-parallel biases (added post-reduce on every
    rank).  Column-parallel weight+bias and row-parallel weights are the
    sharded ones."""
    sharded = set()
    for mod in model.modules():
        if isinstance(mod, ColumnParallelLinear):
            sharded.add(id(mod.weight))
            if mod.bias is not None:
                sharded.add(id(mod.bias))
        elif isinstance(mod, RowParallelLinear):
            sharded.add(id(mod.weight))
    return [p for p in model.parameters() if id(p) not in sharded]


@torch.no_grad()
def tp_resync_replicated(model: nn.Module) -> None:
    """Broadcast replicated params from rank 0.

    Mathematically every rank computes identical gradients for these, but
    the fused kernels accumulate with fp32 atomics (embedding scatter-add,
    attention dQ/dK/dV) in non-deterministic order, so replicated weights
    drift over many steps (round-1 advisor finding #4).  Call periodically
    — Megatron broadcasts its non-sharded params the same way."""
    if _world() == 1:
        return
    for p in tp_replicated_parameters(model):
        dist.broadcast(p.data, src=0)


@torch.no_grad()
def tp_full_state_dict(model: nn.Module):
    """Gather sharded weights back to a full state dict on rank 0."""
    w = _world()
    if w == 1:
        return {k: v.cpu().clone() for k, v in model.state_dict().items()}
    full = {}
    for name, mod in model.named_modules():
        if isinstance(mod, ColumnParallelLinear):
            shards = [torch.empty_like(mod.weight) for _ in range(w)]
            dist.all_gather(shards, mod.weight.data.contiguous())
            full[f"{name}.weight"] = torch.cat(shards, dim=0).cpu()
            if mod.bias is not None:
                bs = [torch.empty_like(mod.bias) for _ in range(w)]
                dist.all_gather(bs, mod.bias.data.contiguous())
                full[f"{name}.bias"] = torch.cat(bs, dim=0).cpu()
        elif isinstance(mod, RowParallelLinear):
            shards = [torch.empty_like(mod.weight) for _ in range(w)]
            dist.all_gather(shards, mod.weight.data.contiguous())
            full[f"{name}.weight"] = torch.cat(shards, dim=1).cpu()
            if mod.bias is not None:
                full[f"{name}.bias"] = mod.bias.data.cpu().clone()
    sd = model.state_dict()
    out = {}
    for k, v in sd.items():
        hit = None
        for fk in full:
            if k == fk or k.endswith("." + fk):
                hit = full[fk]
                break
        out[k] = hit if hit is not None else v.cpu().clone()
    return out if _rank() == 0 else None
