"""ZeRO-3 / FSDP-style sharded data parallelism, built directly on
RCCL-over-xGMI collectives.

MI355X-native replacement for the reference's use of torch FSDP
(``examples/wikitext103/executors/FSDP.py:184-245``): per-unit (transformer
block) flat parameter shards, all-gather before a unit's forward/backward,
reduce-scatter of unit gradients after its backward, one-unit-ahead
all-gather prefetch overlapped with compute, optional
pinned-host shard offload (C9) and per-unit activation checkpointing.

Each unit's parameters live in ONE flat buffer per dtype; ranks own a
1/world slice.  During compute, ``param.data`` views the gathered flat
buffer; outside it, params are dummies and only the shard exists — sized
for 288 GB HBM: on one MI355X even a 70B model's shards fit without
offload, so offload is an autotuned option, not a default.

Collective fallbacks keep the whole path testable on CPU/gloo world_size=2
(gloo lacks reduce_scatter_tensor; emulated with all_reduce + slice).
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional, Sequence

import torch
import torch.distributed as dist
import torch.nn as nn

log = logging.getLogger(__name__)


def _world() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def _rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def _allgather_into(full: torch.Tensor, shard: torch.Tensor, async_op=False):
    if _world() == 1:
        full.copy_(shard)
        return None
    if dist.get_backend() == "gloo":
        chunks = list(full.chunk(_world()))
        return dist.all_gather(chunks, shard, async_op=async_op)
    return dist.all_gather_into_tensor(full, shard, async_op=async_op)


def _reducescatter_into(shard: torch.Tensor, full: torch.Tensor):
    """shard <- sum over ranks of full[own slice].  Averaging is applied by
    the caller."""
    if _world() == 1:
        shard.copy_(full)
        return
    if dist.get_backend() == "gloo":
        dist.all_reduce(full)
        shard.copy_(full.chunk(_world())[_rank()])
        return
    dist.reduce_scatter_tensor(shard, full)


#: one copy stream per device, shared by all offload units — H2D prefetch
#: of unit i+1 overlaps unit i's compute (double-buffered
#: pinned-DRAM <-> HBM streaming on a dedicated HIP copy stream)
_copy_streams: Dict[int, "torch.cuda.Stream"] = {}


def _copy_stream(device) -> Optional["torch.cuda.Stream"]:
    if device.type != "cuda":
        return None
    idx = device.index or 0
    if idx not in _copy_streams:
        _copy_streams[idx] = torch.cuda.Stream(device=device)
    return _copy_streams[idx]


class _Unit:
    """One shard group: a block's parameters flattened per dtype."""

    def __init__(self, idx: int, module: nn.Module, params: List[nn.Parameter],
                 device, offload: bool) -> None:
        self.idx = idx
        self.module = module
        self.params = params
        self.device = device
        self.offload = offload
        world = _world()
        numel = sum(p.numel() for p in params)
        self.pad_numel = (numel + world - 1) // world * world
        self.shard_numel = self.pad_numel // world
        dtype = params[0].dtype
        self.dtype = dtype

        # Build the packed flat once, slice own shard, drop the rest.
        flat = torch.zeros(self.pad_numel, dtype=dtype, device=device)
        offset = 0
        self.offsets: List[int] = []
        self.shapes: List[torch.Size] = []
        self.numels: List[int] = []
        for p in params:
            flat[offset : offset + p.numel()].copy_(p.data.view(-1))
            self.offsets.append(offset)
            self.shapes.append(p.shape)
            self.numels.append(p.numel())
            offset += p.numel()
        r = _rank()
        shard = flat[r * self.shard_numel : (r + 1) * self.shard_numel].clone()
        if offload:
            # allocate the host shard pinned UP FRONT: .to("cpu") followed
            # by .pin_memory() holds two full host copies at once — for a
            # 70B model that is 2 x 140 GB of DRAM and can OOM the host
            # (it killed a GPU box in round 2's config-5 run)
            host = torch.empty(
                shard.shape, dtype=shard.dtype, device="cpu",
                pin_memory=device.type == "cuda",
            )
            host.copy_(shard)
            shard = host
        # The optimizer updates this leaf directly (ZeRO: optimizer state is
        # sharded for free).
        self.shard = nn.Parameter(shard)
        del flat
        # Params become dummies until gathered.
        self._placeholder = torch.empty(0, dtype=dtype, device=device)
        for p in params:
            p.data = self._placeholder
        # ONE persistent gather buffer whose storage is resized 0 <-> full.
        # Autograd's saved tensors (weight views captured by the unit's
        # forward ops) share this storage: resizing to 0 actually frees the
        # HBM after forward, and the backward re-gather refills the SAME
        # storage so those saved views become valid again (the trick torch
        # FSDP uses; merely dropping the Python reference would leave every
        # unit's full buffer pinned until backward — round-1 advisor
        # finding).
        self.full: Optional[torch.Tensor] = None
        self.gathered = False  # gather started (in flight or complete)
        self.work = None
        self.copy_event: Optional["torch.cuda.Event"] = None
        self.grad_pending = 0

    def _materialize_full(self) -> torch.Tensor:
        if self.full is None:
            self.full = torch.empty(
                self.pad_numel, dtype=self.dtype, device=self.device
            )
        elif self.full.untyped_storage().size() == 0:
            self.full.untyped_storage().resize_(
                self.pad_numel * self.full.element_size()
            )
        return self.full

    # -- gather / free -----------------------------------------------------
    def start_gather(self, async_op: bool = False) -> None:
        if self.gathered:
            return
        self.gathered = True
        full = self._materialize_full()
        cs = _copy_stream(self.device) if self.offload else None
        if cs is not None:
            # pinned-host -> HBM on the copy stream: a prefetched unit's
            # upload overlaps the current unit's compute
            full.record_stream(cs)  # allocator: buffer is used on cs
            ev = torch.cuda.Event()
            with torch.cuda.stream(cs):
                shard_dev = self.shard.data.to(self.device, non_blocking=True)
                if _world() == 1:
                    full.copy_(shard_dev)
                    self.work = None
                else:
                    self.work = _allgather_into(
                        full, shard_dev, async_op=async_op
                    )
                ev.record(cs)
            self.copy_event = ev
            return
        shard_dev = self.shard.data
        self.work = _allgather_into(full, shard_dev, async_op=async_op)

    def finish_gather(self) -> None:
        if self.copy_event is not None:
            torch.cuda.current_stream(self.device).wait_event(self.copy_event)
            self.copy_event = None
        if self.work is not None:
            self.work.wait()
            self.work = None
        for p, off, shp, n in zip(self.params, self.offsets, self.shapes, self.numels):
            p.data = self.full[off : off + n].view(shp)

    def free(self) -> None:
        for p in self.params:
            p.data = self._placeholder
        if self.copy_event is not None:
            # never drop a buffer with an in-flight upload
            self.copy_event.synchronize()
            self.copy_event = None
        if self.work is not None:
            self.work.wait()
        if self.full is not None and self.gathered:
            self.full.untyped_storage().resize_(0)
        self.gathered = False
        self.work = None

    # -- gradient reduce-scatter -------------------------------------------
    def reduce_grads(self) -> None:
        full_grad = torch.zeros(
            self.pad_numel, dtype=self.dtype, device=self.device
        )
        for p, off, n in zip(self.params, self.offsets, self.numels):
            if p.grad is not None:
                full_grad[off : off + n].copy_(p.grad.view(-1))
                p.grad = None
        grad_shard = torch.empty(
            self.shard_numel, dtype=self.dtype, device=self.device
        )
        _reducescatter_into(grad_shard, full_grad)
        if _world() > 1:
            grad_shard.div_(_world())
        if self.offload:
            grad_shard = grad_shard.to("cpu")
        if self.shard.grad is None:
            self.shard.grad = grad_shard
        else:
            self.shard.grad.add_(grad_shard)


class Zero3Model(nn.Module):
    """Wrap a model whose child blocks shard independently.

    ``unit_modules`` picks the shard units (default: children of
    ``model.h`` if present, else top-level children); everything not in a
    unit goes into one residual unit (embeddings, final norm, lm_head).
    """

    def __init__(
        self,
        model: nn.Module,
        unit_modules: Optional[Sequence[nn.Module]] = None,
        device: Optional[torch.device] = None,
        offload: bool = False,
        checkpoint_activations: bool = False,
        prefetch: bool = True,
    ) -> None:
        super().__init__()
        self.model = model
        self.device = device or next(model.parameters()).device
        self.checkpoint_activations = checkpoint_activations
        self.prefetch = prefetch and self.device.type == "cuda"

        if unit_modules is None:
            if hasattr(model, "h"):
                unit_modules = list(model.h)
            else:
                unit_modules = [
                    m for m in model.children() if any(p.requires_grad for p in m.parameters())
                ]
        unit_modules = list(unit_modules)

        # rank-0 weights win (mirrors DDP broadcast semantics)
        if _world() > 1:
            with torch.no_grad():
                for p in model.parameters():
                    dist.broadcast(p.data, src=0)

        claimed = set()
        self.units: List[_Unit] = []
        for i, m in enumerate(unit_modules):
            ps = [p for p in m.parameters() if p.requires_grad]
            self.units.append(_Unit(i, m, ps, self.device, offload))
            claimed |= {id(p) for p in ps}
        rest = [
            p for p in model.parameters() if p.requires_grad and id(p) not in claimed
        ]
        if rest:
            self.units.append(
                _Unit(len(self.units), model, rest, self.device, offload)
            )
        self._unit_of_module: Dict[int, _Unit] = {
            id(u.module): u for u in self.units
        }
        self._install_hooks()

    # -- hooks -------------------------------------------------------------
    @staticmethod
    def _triggers(u: _Unit) -> List[nn.Module]:
        """Forward-invoked modules of a unit.  A container unit (e.g. a
        ModuleList grouping several blocks for the spill executor's
        partitions) never runs its own forward — hook its children."""
        m = u.module
        if isinstance(m, (nn.ModuleList, nn.ModuleDict)):
            return list(m.children())
        return [m]

    def _install_hooks(self) -> None:
        for u in self.units:
            if u.module is self.model:
                continue
            triggers = self._triggers(u)
            for t in triggers:
                t.register_forward_pre_hook(self._fwd_pre(u))
                t.register_full_backward_pre_hook(self._bwd_pre(u))
            triggers[-1].register_forward_hook(self._fwd_post(u))
            for p in u.params:
                p.register_post_accumulate_grad_hook(self._grad_hook(u))

    def _fwd_pre(self, u: _Unit):
        def hook(module, args):
            u.start_gather()  # no-op if already gathered
            u.finish_gather()
            # prefetch next unit's gather onto the fabric
            if self.prefetch and u.idx + 1 < len(self.units):
                nxt = self.units[u.idx + 1]
                if nxt.module is not self.model:
                    nxt.start_gather(async_op=True)
        return hook

    def _fwd_post(self, u: _Unit):
        def hook(module, args, output):
            u.free()  # re-gathered at backward by _bwd_pre
        return hook

    def _bwd_pre(self, u: _Unit):
        def hook(module, grad_output):
            u.start_gather()
            u.finish_gather()
            if u.grad_pending == 0:
                u.grad_pending = len(u.params)
            if self.prefetch and u.idx - 1 >= 0:
                prv = self.units[u.idx - 1]
                if prv.module is not self.model:
                    prv.start_gather(async_op=True)
        return hook

    def _grad_hook(self, u: _Unit):
        def hook(p):
            u.grad_pending -= 1
            if u.grad_pending == 0:
                u.reduce_grads()
                u.free()
        return hook

    # -- residual unit (embeddings/head) is gathered for the whole step ----
    def forward(self, *args, **kwargs):
        res = self.units[-1]
        if res.module is self.model:
            res.start_gather()
            res.finish_gather()
            res.grad_pending = len(res.params)
        if self.checkpoint_activations:
            out = self._forward_with_checkpointing(*args, **kwargs)
        else:
            out = self.model(*args, **kwargs)
        return out

    def _forward_with_checkpointing(self, *args, **kwargs):
        """Per-unit activation checkpointing for models exposing .h blocks
        (reference FSDP.py:214-217 wraps blocks the same way)."""
        from torch.utils.checkpoint import checkpoint

        m = self.model
        if not hasattr(m, "wte") or not hasattr(m, "h"):
            return m(*args, **kwargs)
        x = args[0]
        h = m.wte(x)
        if hasattr(m, "wpe"):
            pos = torch.arange(x.shape[1], device=x.device)
            h = h + m.wpe(pos)[None]
        for blk in m.h:
            h = checkpoint(blk, h, use_reentrant=False)
        h = m.ln_f(h)
        return m.lm_head(h)

    def grad_sync(self) -> None:
        """Finish the step: reduce the residual unit's grads."""
        res = self.units[-1]
        if res.module is self.model and res.gathered:
            res.reduce_grads()
            res.free()

    def sharded_parameters(self) -> List[nn.Parameter]:
        """Flat shard leaves for the optimizer (ZeRO: optimizer state is
        sharded with them)."""
        return [u.shard for u in self.units]

    def zero_grad_shards(self) -> None:
        for u in self.units:
            u.shard.grad = None

    # -- checkpointing -----------------------------------------------------
    @torch.no_grad()
    def full_state_dict(self) -> Optional[Dict[str, torch.Tensor]]:
        """Gather a FULL state dict onto rank 0 / CPU (reference
        FSDP.py:239-244's FULL_STATE_DICT + cpu-offload save)."""
        for u in self.units:
            u.start_gather()
            u.finish_gather()
        sd = None
        if _rank() == 0:
            sd = {k: v.detach().cpu().clone() for k, v in self.model.state_dict().items()}
        for u in self.units:
            u.free()
        return sd
