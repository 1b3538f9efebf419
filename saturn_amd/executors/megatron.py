"""Tensor-parallel (Megatron-style) executor.

The reference declares this technique but never implements it
(``Techniques.MEGATRON``, Strategy.py:34 — SURVEY §2.2).  Here it is a
selectable library member: per-rank column/row-parallel shards via
``saturn_amd.parallel.tensor``, one process per GPU over RCCL, every rank
consuming the same batches (pure TP).
"""

from __future__ import annotations

from timeit import default_timer as timer
from typing import Any, Dict, List, Optional, Tuple

from saturn_amd.core.technique import BaseTechnique
from saturn_amd.executors.launch import (
    destroy_process_group,
    gang_spawn,
    init_process_group,
)

TRIAL_BATCHES = 3


def _tp_worker(rank: int, world: int, task, tid: int, batch_count: int,
               params: Optional[Dict[str, Any]], trial: bool):
    import torch
    import torch.distributed as dist

    from saturn_amd.executors.ddp import _make_optimizer
    from saturn_amd.parallel.tensor import (
        tp_full_state_dict,
        tp_resync_replicated,
        tp_shard_model,
    )

    backend = init_process_group(rank, world)
    try:
        device = (
            torch.device("cuda", rank) if backend == "nccl" else torch.device("cpu")
        )
        dtype = torch.bfloat16 if backend == "nccl" else torch.float32
        from saturn_amd.executors.ddp import build_model_on

        model = build_model_on(task, device, dtype)
        model.train()
        if world > 1:
            with torch.no_grad():
                for p in model.parameters():
                    dist.broadcast(p.data, src=0)
        model = tp_shard_model(model)
        optimizer = _make_optimizer(task, model)

        # Per-rank optimizer shard resume (mirrors FSDPExecutor: the
        # reference loses moments at every interval; TP shards
        # are rank-local so each rank reloads its own state when the solver
        # kept the same world size)
        import os as _os

        opt_path = _os.path.join(
            task.save_dir, f"{task.name}.tpopt.w{world}.r{rank}.pt"
        )
        if not trial and _os.path.isfile(opt_path):
            try:
                optimizer.load_state_dict(
                    torch.load(opt_path, map_location="cpu", weights_only=False)
                )
            except Exception:
                pass  # layout changed; fresh moments

        it = task.get_iterator() if not trial else task.get_fresh_iterator()

        def next_batch():
            nonlocal it
            try:
                return next(it)
            except StopIteration:
                it = task.get_fresh_iterator()
                return next(it)

        def step(batch):
            x, y = batch
            x = x.to(device, non_blocking=True)
            y = y.to(device, non_blocking=True)
            if x.is_floating_point():
                x = x.to(dtype)
            loss = task.loss_function(model(x), y)
            loss.backward()
            optimizer.step()
            optimizer.zero_grad(set_to_none=True)

        if trial:
            step(next_batch())
            if device.type == "cuda":
                torch.cuda.synchronize()
                torch.cuda.reset_peak_memory_stats()
            t0 = timer()
            for _ in range(TRIAL_BATCHES - 1):
                step(next_batch())
            if device.type == "cuda":
                torch.cuda.synchronize()
            hbm = (
                torch.cuda.max_memory_allocated() / 2**30
                if device.type == "cuda"
                else 0.0
            )
            return ((timer() - t0) / (TRIAL_BATCHES - 1), hbm)
        RESYNC_EVERY = 64  # bound replicated-param drift from fp32 atomics
        for i in range(batch_count):
            step(next_batch())
            if (i + 1) % RESYNC_EVERY == 0:
                tp_resync_replicated(model)
        if device.type == "cuda":
            torch.cuda.synchronize()
        tp_resync_replicated(model)  # checkpoint from a rank-consistent state
        sd = tp_full_state_dict(model)
        if rank == 0 and sd is not None:
            task.save_checkpoint(sd, None)
        if _os.environ.get("SATURN_SKIP_CKPT") != "1":
            tmp = opt_path + ".tmp"
            torch.save(optimizer.state_dict(), tmp)
            _os.replace(tmp, opt_path)
        if world > 1:
            dist.barrier()
        return None
    finally:
        destroy_process_group()


class MegatronExecutor(BaseTechnique):
    """Tensor parallelism (column/row-parallel linears, head sharding)."""

    name = "megatron"

    @staticmethod
    def execute(task, gpus: List[int], tid: int, batch_count: int) -> None:
        params = (
            task.selected_strategy.parameters
            if task.selected_strategy is not None
            else {}
        )
        gang_spawn(_tp_worker, len(gpus), tid, task, tid, batch_count, params, False)

    @staticmethod
    def search(
        task, gpus: List[int], tid: int
    ) -> Tuple[Optional[Dict[str, Any]], float]:
        if len(gpus) < 2:
            return None, float("inf")  # TP over 1 GPU is plain training
        try:
            out = gang_spawn(
                _tp_worker, len(gpus), tid, task, tid, TRIAL_BATCHES,
                {"tp": len(gpus)}, True,
            )
        except Exception:
            return None, float("inf")
        if out is None:
            return None, float("inf")
        bt, hbm = out
        return dict(tp=len(gpus), hbm_peak_gb=round(hbm, 2)), bt
