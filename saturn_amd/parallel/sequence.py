"""Ulysses-style sequence parallelism (SP) over RCCL/xGMI.

Beyond the reference's scope (no sequence/context parallelism
exists there; the survey notes SP "slots into the Library as one more UDP"
— this module is that executor's engine):

- every rank holds the full batch's token ids but computes on a contiguous
  1/P slice of the sequence; all pointwise-over-token layers (embeddings,
  norms, MLPs, projections, loss) run on the local shard untouched;
- attention needs the full sequence, so q/k/v ride an all-to-all that
  trades the sequence shard for a head shard ([B, T/P, H, D] ->
  [B, T, H/P, D]), the fused flash kernel runs with exact causal masking,
  and the output rides the inverse all-to-all;
- weight gradients differ per rank (different tokens), so the executor
  wraps the model in the bucketed DDP engine for the gradient all-reduce.

``sp_region(world)`` arms the dispatch inside
``saturn_amd.ops.functional.causal_attention`` so every model in the zoo is
SP-capable without modification.  The all-to-all is autograd-transparent
(backward = inverse all-to-all); on gloo (CPU tests) it is emulated with
all_gather + slicing.
"""

from __future__ import annotations

import contextlib
from typing import Optional

import torch
import torch.distributed as dist

_STATE = {"world": 1, "rank": 0, "group": None}


def sp_world() -> int:
    return _STATE["world"]


@contextlib.contextmanager
def sp_region(world: int, rank: int, group=None):
    """Enable sequence-parallel attention dispatch inside the context."""
    prev = dict(_STATE)
    _STATE.update(world=world, rank=rank, group=group)
    try:
        yield
    finally:
        _STATE.update(prev)


def _all_to_all_4d(x: torch.Tensor, scatter_dim: int, gather_dim: int,
                   world: int, group) -> torch.Tensor:
    """all_to_all over equal chunks: scatter x along scatter_dim, gather
    along gather_dim.  gloo lacks all_to_all -> all_gather + local slice."""
    if dist.get_backend(group) == "gloo":
        pieces = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(pieces, x.contiguous(), group=group)
        rank = dist.get_rank(group)
        mine = [
            p.chunk(world, dim=scatter_dim)[rank] for p in pieces
        ]
        return torch.cat(mine, dim=gather_dim).contiguous()
    send = torch.cat(
        [c.contiguous() for c in x.chunk(world, dim=scatter_dim)], dim=0
    )
    recv = torch.empty_like(send)
    dist.all_to_all_single(recv, send, group=group)
    parts = recv.chunk(world, dim=0)
    return torch.cat(parts, dim=gather_dim).contiguous()


class _SeqToHead(torch.autograd.Function):
    """[B, H, T/P, D] -> [B, H/P, T, D] (and inverse in backward)."""

    @staticmethod
    def forward(ctx, x, world, group):
        ctx.world = world
        ctx.group = group
        return _all_to_all_4d(x, scatter_dim=1, gather_dim=2, world=world,
                              group=group)

    @staticmethod
    def backward(ctx, g):
        return (
            _all_to_all_4d(g.contiguous(), scatter_dim=2, gather_dim=1,
                           world=ctx.world, group=ctx.group),
            None,
            None,
        )


class _HeadToSeq(torch.autograd.Function):
    """[B, H/P, T, D] -> [B, H, T/P, D] (and inverse in backward)."""

    @staticmethod
    def forward(ctx, x, world, group):
        ctx.world = world
        ctx.group = group
        return _all_to_all_4d(x, scatter_dim=2, gather_dim=1, world=world,
                              group=group)

    @staticmethod
    def backward(ctx, g):
        return (
            _all_to_all_4d(g.contiguous(), scatter_dim=1, gather_dim=2,
                           world=ctx.world, group=ctx.group),
            None,
            None,
        )


def sp_attention(q, k, v, attention_fn):
    """The Ulysses exchange around a full-sequence attention core.
    q, k, v: [B, H, T_local, D]."""
    world, group = _STATE["world"], _STATE["group"]
    if world == 1:
        return attention_fn(q, k, v)
    assert q.shape[1] % world == 0, "heads must divide SP world"
    qf = _SeqToHead.apply(q, world, group)
    kf = _SeqToHead.apply(k, world, group)
    vf = _SeqToHead.apply(v, world, group)
    of = attention_fn(qf, kf, vf)
    return _HeadToSeq.apply(of, world, group)
