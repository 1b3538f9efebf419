"""DDP executor: data-parallel training, one process per GPU over RCCL.

Capability parity with the reference example UDP
(``examples/wikitext103/executors/DDP.py:41-182``) with its selection bug
fixed: ``search`` returns a real parameter dict on success (the reference
returned ``(None, rt)`` and was therefore never selectable,
DDP.py:71-72 vs PerformanceEvaluator.py:110).

The gradient synchronization is this framework's own bucketed flat-buffer
engine (``saturn_amd.parallel.ddp.BucketedDDP``) on RCCL-over-xGMI, and the
optimizer step is the fused multi-tensor HIP kernel when running on GPU
(``saturn_amd.ops``).
"""

from __future__ import annotations

import os
from timeit import default_timer as timer
from typing import Any, Dict, List, Optional, Tuple

from saturn_amd.core.technique import BaseTechnique
from saturn_amd.executors.launch import (
    destroy_process_group,
    gang_spawn,
    init_process_group,
)

TRIAL_BATCHES = 3  # 1 warmup + 2 timed (reference used 2: DDP.py:43)


def build_model_on(task, device, dtype):
    """Build the task's model directly on ``device`` in ``dtype``.

    Host-side fp32 init + H2D copy of a large model dominates short trial
    cells, and a 70B-class model can't even transit fp32-on-device
    (280 GB weights + the bf16 copy during .to()).  Building under the
    device context with the default dtype set to the compute dtype
    materializes weights once, on-device, at final precision.
    """
    import torch

    if device.type != "cuda":
        return task.get_model().to(device=device, dtype=dtype)
    prev = torch.get_default_dtype()
    torch.set_default_dtype(dtype)
    try:
        with device:
            model = task.get_model()
    finally:
        torch.set_default_dtype(prev)
    # .to() is a no-op for params already in (device, dtype); it still
    # sweeps stray buffers a factory may have made elsewhere
    return model.to(device=device, dtype=dtype)


def _make_optimizer(task, model):
    """Task-specified optimizer, defaulting to the framework's fused SGD on
    GPU / torch SGD on CPU."""
    import torch

    params = list(model.parameters())
    cls = task.hparams.optimizer_cls
    if cls is None:
        from saturn_amd.ops.optim import FusedSGD

        return FusedSGD(params, lr=task.hparams.lr)
    return cls(params, lr=task.hparams.lr)


def _ddp_worker(
    rank: int,
    world: int,
    task,
    tid: int,
    batch_count: int,
    params: Optional[Dict[str, Any]],
    trial: bool,
):
    """One rank of the DDP gang.  Returns (rank 0) the measured per-batch
    seconds when ``trial`` else None."""
    import torch

    from saturn_amd.parallel.ddp import BucketedDDP

    backend = init_process_group(rank, world)
    try:
        device = torch.device("cuda", rank) if backend == "nccl" else torch.device("cpu")
        dtype = torch.bfloat16 if backend == "nccl" else torch.float32

        model = build_model_on(task, device, dtype)
        model.train()
        bucket_mb = float((params or {}).get("bucket_mb", 64.0))
        comm = None
        if backend == "nccl" and os.environ.get("SATURN_NATIVE_COMM", "0") == "1":
            from saturn_amd.comm import create_comm

            comm = create_comm(rank, world)
        ddp = BucketedDDP(model, bucket_mb=bucket_mb, comm=comm)
        optimizer = _make_optimizer(task, model)

        ckpt = task.load_checkpoint()
        if ckpt is not None and ckpt.get("optimizer") is not None and not trial:
            try:
                optimizer.load_state_dict(ckpt["optimizer"])
            except Exception:
                pass  # optimizer class may have changed between intervals

        it = task.get_iterator() if not trial else task.get_fresh_iterator()

        last_loss = [0.0]

        def step(batch) -> None:
            x, y = batch
            # Shard the GLOBAL batch across ranks (the reference's
            # DistributedSampler semantics, DDP.py:117-144): one step
            # consumes exactly one dataloader batch regardless of world
            # size, so the parent's cursor advance (reconfigure) and the
            # solver's runtime model (per-step time drops with more GPUs)
            # both stay exact.  Round-robin whole batches — the round-1
            # design — consumed world× batches per step, which the
            # accounting missed (advisor finding #5).  A batch smaller
            # than the world is replicated (identical grads; average is a
            # no-op); a non-divisible batch gives ranks shards differing
            # by one row, biasing the grad average by O(1/n) — same as
            # torch's DistributedSampler without padding.
            n = x.shape[0]
            if world > 1 and n >= world:
                lo = rank * n // world
                hi = (rank + 1) * n // world
                x, y = x[lo:hi], y[lo:hi]
            x = x.to(device, non_blocking=True)
            y = y.to(device, non_blocking=True)
            if x.is_floating_point():
                x = x.to(dtype)
            out = ddp(x)
            loss = task.loss_function(out, y)
            loss.backward()
            ddp.grad_sync()
            optimizer.step()
            ddp.zero_grad_buffers()
            last_loss[0] = loss.detach()

        def next_batch():
            nonlocal it
            try:
                return next(it)
            except StopIteration:
                it = task.get_fresh_iterator()
                return next(it)

        result = None
        if trial:
            n_timed = batch_count - 1
            step(next_batch())  # warmup
            if device.type == "cuda":
                torch.cuda.synchronize()
                torch.cuda.reset_peak_memory_stats()
            t0 = timer()
            for _ in range(n_timed):
                step(next_batch())
            if device.type == "cuda":
                torch.cuda.synchronize()
            bt = (timer() - t0) / max(1, n_timed)
            hbm = (
                torch.cuda.max_memory_allocated() / 2**30
                if device.type == "cuda"
                else 0.0
            )
            result = (bt, hbm)
        else:
            for _ in range(batch_count):
                step(next_batch())
            if device.type == "cuda":
                torch.cuda.synchronize()
            if rank == 0:
                import logging

                logging.getLogger(__name__).info(
                    "task %s: %d batches done, loss %.4f",
                    task.name, batch_count, float(last_loss[0]),
                )
                task.save_checkpoint(model, optimizer)
            import torch.distributed as dist

            if world > 1:
                dist.barrier()
        return result
    finally:
        destroy_process_group()


class DDPExecutor(BaseTechnique):
    """Data parallelism via bucketed RCCL all-reduce."""

    name = "ddp"

    @staticmethod
    def execute(task, gpus: List[int], tid: int, batch_count: int) -> None:
        params = (
            task.selected_strategy.parameters
            if task.selected_strategy is not None
            else None
        )
        gang_spawn(
            _ddp_worker,
            len(gpus),
            tid,
            task,
            tid,
            batch_count,
            params,
            False,
        )

    @staticmethod
    def search(
        task, gpus: List[int], tid: int
    ) -> Tuple[Optional[Dict[str, Any]], float]:
        """Time a short trial; tune the bucket size over a small grid.

        Returns ({"bucket_mb": best}, seconds_per_batch) or (None, inf) when
        the trial cannot run (e.g. OOM)."""
        world = len(gpus)
        candidates = [64.0] if world == 1 else [32.0, 128.0]
        best: Tuple[Optional[Dict[str, Any]], float] = (None, float("inf"))
        for mb in candidates:
            try:
                out = gang_spawn(
                    _ddp_worker,
                    world,
                    tid,
                    task,
                    tid,
                    TRIAL_BATCHES,
                    {"bucket_mb": mb},
                    True,
                )
            except Exception:
                continue
            if out is None:
                continue
            bt, hbm = out
            if bt < best[1]:
                # measured HBM high-water rides in the params dict — the
                # rocm-smi-grade memory signal the reference approximates
                # with exception-string matching
                best = ({"bucket_mb": mb, "hbm_peak_gb": round(hbm, 2)}, bt)
        return best
