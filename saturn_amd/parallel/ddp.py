"""Bucketed data-parallel gradient synchronization for xGMI.

MI355X-native replacement for ``torch.nn.parallel.DistributedDataParallel``
as used by the reference's DDP executor (DDP.py:90,155): the hard systems
work the reference delegated to torch's C++ reducer is done here explicitly,
laid out for the MI355X node fabric:

- gradients accumulate directly into one flat per-bucket buffer
  (``param.grad`` is a view into it), so there is no pack/unpack kernel at
  all — the bucket is RCCL-ready the moment its last grad lands;
- buckets are built in reverse parameter order (backward completes roughly
  in reverse) and all-reduced asynchronously as they fill, overlapping the
  remaining backward;
- the default bucket size targets the 7-link xGMI fabric: RCCL pipelines
  multiple in-flight ring all-reduces across links, so several mid-size
  buckets in flight beat one giant one.  Default 64 MiB.

Runs on ``nccl`` (= RCCL) on GPU and on ``gloo`` for the CPU test suite —
same code path, same hooks.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


class _Bucket:
    def __init__(self, params: List[torch.nn.Parameter], device, dtype) -> None:
        self.params = params
        numels = [p.numel() for p in params]
        self.flat = torch.zeros(sum(numels), device=device, dtype=dtype)
        # Carve grad views out of the flat buffer.
        offset = 0
        self.views = []
        for p, n in zip(params, numels):
            self.views.append(self.flat[offset : offset + n].view_as(p))
            offset += n
        self.pending = 0
        self.work: Optional[dist.Work] = None
        self.launched = False  # native-comm path has no Work handle

    def attach_grads(self) -> None:
        for p, v in zip(self.params, self.views):
            p.grad = v

    def reset(self) -> None:
        self.pending = len(self.params)
        self.work = None
        self.launched = False


class BucketedDDP(torch.nn.Module):
    """Wrap a module; forward passes through; gradients are averaged across
    the process group during backward.  Call :meth:`grad_sync` after
    ``loss.backward()`` and before ``optimizer.step()``."""

    def __init__(
        self,
        module: torch.nn.Module,
        bucket_mb: float = 64.0,
        process_group=None,
        grad_dtype: Optional[torch.dtype] = None,
        comm=None,
        exclude=None,
        use_buckets: Optional[bool] = None,
    ) -> None:
        """``comm``: an optional saturn_amd native RcclComm; when given,
        bucket all-reduces run on its dedicated HIP stream (ncclAvg)
        instead of torch.distributed's process group.

        ``exclude``: parameters to leave out of broadcast + bucketed
        all-reduce (expert-parallel shards own their grads; see
        parallel/expert.py)."""
        super().__init__()
        self.module = module
        self.pg = process_group
        self.comm = comm
        if comm is not None:
            self.world = comm.world
        else:
            self.world = (
                dist.get_world_size(process_group) if dist.is_initialized() else 1
            )

        # Broadcast initial parameters from rank 0 (reference relies on the
        # DDP ctor for this, DDP.py:90).
        skip = {id(p) for p in (exclude or ())}
        if self.world > 1:
            with torch.no_grad():
                if comm is not None:
                    for p in module.parameters():
                        if id(p) not in skip:
                            comm.broadcast(p.data, 0)
                    comm.join()
                else:
                    for p in module.parameters():
                        if id(p) not in skip:
                            dist.broadcast(p.data, src=0, group=self.pg)

        params = [
            p for p in module.parameters()
            if p.requires_grad and id(p) not in skip
        ]
        self.buckets: List[_Bucket] = []
        self._param_bucket = {}
        if use_buckets is None:
            use_buckets = self.world > 1
        if not use_buckets:
            # No comm -> no buckets.  Grad-as-bucket-view costs ~4 bytes of
            # pure bookkeeping HBM traffic per gradient element per step
            # (zero the flat + autograd accumulate-into-view); at world 1 the flat
            # buffer serves nothing, so let autograd assign fresh grads and
            # zero with set_to_none.  (Measured: ~8 ms/step of FillFunctor +
            # CUDAFunctor_add on GPT-J-6B, profiles/r02_baseline_kernels.txt)
            return
        bucket_bytes = int(bucket_mb * 1024 * 1024)
        cur: List[torch.nn.Parameter] = []
        cur_bytes = 0
        for p in reversed(params):  # backward finishes roughly in reverse
            if cur and (
                cur_bytes + p.numel() * p.element_size() > bucket_bytes
                or p.dtype != cur[0].dtype
                or p.device != cur[0].device
            ):
                self._seal(cur, grad_dtype)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += p.numel() * p.element_size()
        if cur:
            self._seal(cur, grad_dtype)

        for b in self.buckets:
            b.attach_grads()
            b.reset()
            for p in b.params:
                self._param_bucket[p] = b
                p.register_post_accumulate_grad_hook(self._hook)

    def _seal(self, params: List[torch.nn.Parameter], grad_dtype) -> None:
        p0 = params[0]
        self.buckets.append(
            _Bucket(params, p0.device, grad_dtype or p0.dtype)
        )

    def _launch(self, b: _Bucket) -> None:
        if self.comm is not None:
            # enqueued on the native engine's comm stream, fenced
            # against the producing compute stream — overlaps backward
            self.comm.all_reduce(b.flat, True)
            b.launched = True
        else:
            b.work = dist.all_reduce(b.flat, async_op=True, group=self.pg)

    def _hook(self, p: torch.nn.Parameter) -> None:
        b = self._param_bucket[p]
        b.pending -= 1
        if b.pending == 0 and self.world > 1:
            self._launch(b)

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def grad_sync(self) -> None:
        """Wait for in-flight bucket all-reduces and average.

        A bucket whose params received no grad this step (``pending > 0``,
        e.g. an unused head) is still all-reduced here — its flat buffer
        holds zeros for the missing params, so the average is correct as
        long as the *set* of grad-less params matches across ranks (it is
        structural: same model, same step).  Launching here instead of
        silently scaling an un-reduced bucket fixes the divergence the
        round-1 advisor flagged.
        """
        if self.world > 1:
            for b in self.buckets:
                launched = (
                    b.launched if self.comm is not None else b.work is not None
                )
                if not launched:
                    self._launch(b)
        if self.comm is not None:
            if self.world > 1:
                self.comm.join()  # compute stream waits the comm stream
            for b in self.buckets:
                b.reset()
            return
        inv = 1.0 / self.world
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
            if self.world > 1:
                b.flat.mul_(inv)
            b.reset()

    def zero_grad_buffers(self, set_to_none: bool = False) -> None:
        """Zero the flat grad buffers (grads are views; never set to None).
        At world 1 there are no buckets: plain set_to_none zeroing."""
        if not self.buckets:
            self.module.zero_grad(set_to_none=True)
            return
        for b in self.buckets:
            b.flat.zero_()
            b.attach_grads()
            b.reset()
