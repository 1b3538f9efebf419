"""Expert parallelism: MoE experts sharded across ranks, tokens dispatched
with variable-split all-to-all over RCCL (xGMI is all-pairs point-to-point,
so the dense token exchange maps onto direct links — no staging hop).

Scheme (exact, capacity-free — no token dropping):

1. every rank routes its local tokens (router weights are replicated and
   DDP-synced like any shared parameter);
2. the (token, slot) pairs are sorted by destination expert and exchanged
   with one ``all_to_all`` keyed by per-(rank, expert) counts (counts travel
   first as one small fixed-shape all_gather);
3. each rank applies its owned experts to the tokens it received, grouped
   contiguously per expert — on GPU each group is one hipBLASLt GEMM batch
   through the fused SwiGLU, exactly like a dense FFN;
4. a mirror all_to_all returns the expert outputs; the gate weights are
   applied on the *source* rank so the routing gradient never crosses the
   wire.

Gradient semantics (``ep_scale_expert_grads``): each rank backprops its
local-mean loss; shared params are all-reduce-averaged by BucketedDDP, and
expert grads — which already accumulate contributions from every global
token routed to them via the all-to-all backward — are divided by the world
size so both match single-process training on the global batch (the EP
equivalence test pins this exactly).

On gloo (CPU tests) the variable all-to-all is emulated with
``batch_isend_irecv``; NCCL/RCCL uses ``all_to_all_single``.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from saturn_amd.models.mixtral import MoEMLP, route


def _a2a_impl(x: torch.Tensor, out_splits: List[int], in_splits: List[int],
              group) -> torch.Tensor:
    """Variable-split all-to-all of rows of ``x`` (no autograd)."""
    out = x.new_empty(int(sum(out_splits)), *x.shape[1:])
    backend = dist.get_backend(group)
    if backend == "nccl":
        dist.all_to_all_single(out, x.contiguous(), out_splits, in_splits,
                               group=group)
        return out
    # gloo: pairwise sends/recvs (send my slice r, receive theirs)
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    in_off = [0]
    for c in in_splits:
        in_off.append(in_off[-1] + c)
    out_off = [0]
    for c in out_splits:
        out_off.append(out_off[-1] + c)
    out[out_off[rank]:out_off[rank + 1]] = x[in_off[rank]:in_off[rank + 1]]
    ops = []
    for r in range(world):
        if r == rank:
            continue
        if in_splits[r]:
            ops.append(dist.P2POp(dist.isend,
                                  x[in_off[r]:in_off[r + 1]].contiguous(), r,
                                  group=group))
        if out_splits[r]:
            ops.append(dist.P2POp(dist.irecv, out[out_off[r]:out_off[r + 1]],
                                  r, group=group))
    if ops:
        for w in dist.batch_isend_irecv(ops):
            w.wait()
    return out


class _AllToAllV(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.splits = (in_splits, out_splits)
        ctx.group = group
        return _a2a_impl(x, out_splits, in_splits, group)

    @staticmethod
    def backward(ctx, grad):
        in_splits, out_splits = ctx.splits
        return (
            _a2a_impl(grad.contiguous(), in_splits, out_splits, ctx.group),
            None, None, None,
        )


class DistributedMoE(nn.Module):
    """EP wrapper for one :class:`MoEMLP`: this rank keeps experts
    ``[rank * n/world, (rank+1) * n/world)``; the rest are dropped (their
    memory is freed — the point of EP)."""

    def __init__(self, moe: MoEMLP, group=None):
        super().__init__()
        self.group = group
        self.world = dist.get_world_size(group) if dist.is_initialized() else 1
        self.rank = dist.get_rank(group) if dist.is_initialized() else 0
        self.n_expert = moe.n_expert
        self.top_k = moe.top_k
        if self.n_expert % self.world != 0:
            raise ValueError(
                f"n_expert {self.n_expert} not divisible by world {self.world}"
            )
        self.per_rank = self.n_expert // self.world
        self.e0 = self.rank * self.per_rank
        self.router = moe.router
        self.local_experts = nn.ModuleList(
            moe.experts[self.e0 + i] for i in range(self.per_rank)
        )

    def forward(self, x):
        B, T, E = x.shape
        xf = x.reshape(-1, E)
        weights, topi = route(self.router, xf, self.top_k)
        if self.world == 1:
            out = torch.zeros_like(xf)
            for i, expert in enumerate(self.local_experts):
                tok, slot = (topi == i).nonzero(as_tuple=True)
                if tok.numel():
                    out = out.index_add(
                        0, tok, expert(xf[tok]) * weights[tok, slot].unsqueeze(-1)
                    )
            return out.reshape(B, T, E)

        flat_e = topi.reshape(-1)  # pair p = token p // k, slot p % k
        order = torch.argsort(flat_e, stable=True)
        pair_tok = order // self.top_k
        counts = torch.bincount(flat_e, minlength=self.n_expert)
        in_splits = counts.reshape(self.world, self.per_rank).sum(-1).tolist()

        # per-(source rank, expert) counts: one fixed-shape all_gather
        all_counts = [torch.empty_like(counts) for _ in range(self.world)]
        dist.all_gather(all_counts, counts, group=self.group)
        mine = torch.stack(all_counts)[:, self.e0:self.e0 + self.per_rank]
        out_splits = mine.sum(-1).tolist()

        sent = _AllToAllV.apply(xf[pair_tok], out_splits, in_splits, self.group)

        # received rows arrive grouped by source rank, each group sorted by
        # expert id — regroup contiguously per owned expert
        offs = [0]
        for r in range(self.world):
            offs.append(offs[-1] + int(out_splits[r]))
        pieces = []
        pos = 0
        back_index = torch.empty(sent.shape[0], dtype=torch.long,
                                 device=sent.device)
        for i in range(self.per_rank):
            for r in range(self.world):
                lo = offs[r] + int(mine[r, :i].sum())
                hi = lo + int(mine[r, i])
                n = hi - lo
                if n == 0:
                    continue
                pieces.append(self.local_experts[i](sent[lo:hi]))
                back_index[lo:hi] = torch.arange(pos, pos + n,
                                                 device=sent.device)
                pos += n
        y = (torch.cat(pieces).index_select(0, back_index)
             if pieces else sent)

        back = _AllToAllV.apply(y, in_splits, out_splits, self.group)
        w = weights.reshape(-1)[order].unsqueeze(-1).to(back.dtype)
        out = torch.zeros_like(xf).index_add(0, pair_tok, back * w)
        return out.reshape(B, T, E)


def ep_shard_model(model: nn.Module, group=None) -> nn.Module:
    """Replace every MoEMLP in ``model`` with its :class:`DistributedMoE`
    shard (non-owned experts are freed)."""
    for mod in model.modules():
        for name, child in list(mod.named_children()):
            if isinstance(child, MoEMLP):
                setattr(mod, name, DistributedMoE(child, group))
    return model


def ep_expert_parameters(model: nn.Module) -> List[torch.nn.Parameter]:
    out = []
    for mod in model.modules():
        if isinstance(mod, DistributedMoE):
            for e in mod.local_experts:
                out.extend(e.parameters())
    return out


def ep_scale_expert_grads(model: nn.Module) -> None:
    """Divide local expert grads by world (see module docstring)."""
    world = dist.get_world_size() if dist.is_initialized() else 1
    if world == 1:
        return
    with torch.no_grad():
        for p in ep_expert_parameters(model):
            if p.grad is not None:
                p.grad.div_(world)


def ep_full_state_dict(model: nn.Module) -> Optional[dict]:
    """Reassemble the full (unsharded) state dict on rank 0: shared params
    from rank 0 plus every expert gathered from its owner, keyed exactly as
    the original :class:`MoEMLP` module tree."""
    world = dist.get_world_size() if dist.is_initialized() else 1
    rank = dist.get_rank() if dist.is_initialized() else 0
    sd = {}
    for name, mod in model.named_modules():
        if not isinstance(mod, DistributedMoE):
            continue
        for g in range(mod.n_expert):
            owner = g // mod.per_rank
            local = g - owner * mod.per_rank
            for pn in ("gate_proj", "up_proj", "down_proj"):
                key = f"{name}.experts.{g}.{pn}.weight"
                if owner == rank:
                    t = getattr(mod.local_experts[local], pn).weight.data
                else:
                    t = None
                if world > 1:
                    shape_src = (
                        getattr(mod.local_experts[local % mod.per_rank], pn)
                        .weight.shape
                    )
                    buf = (
                        t.contiguous()
                        if t is not None
                        else torch.empty(
                            shape_src,
                            dtype=mod.router.weight.dtype,
                            device=mod.router.weight.device,
                        )
                    )
                    dist.broadcast(buf, src=owner)
                    if rank == 0:
                        sd[key] = buf.cpu()
                elif rank == 0:
                    sd[key] = t.cpu()
    if rank != 0:
        return None
    full = {
        k: v.cpu()
        for k, v in model.state_dict().items()
        if "local_experts" not in k
    }
    # DistributedMoE stores the router under the same relative name
    full.update(sd)
    return full
