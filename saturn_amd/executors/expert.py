"""Expert-parallel executor (EP × DP).

Beyond the reference's technique set (SURVEY §2.2 lists EP as absent
upstream).  One process per GPU over RCCL: experts are sharded across the
gang (``parallel/expert.py`` all-to-all token dispatch), the non-expert
parameters are replicated and synced by BucketedDDP, and each rank consumes
its own shard of the batch — so the gang is simultaneously data-parallel
over tokens and expert-parallel over FFN weights, the standard MoE layout.

Checkpoint: rank 0 saves the reassembled dense state dict
(``ep_full_state_dict``) to the standard ``<name>.pt``, so any other
technique (or a plain single-GPU run) can resume the same task; each rank
additionally persists its optimizer shard like the FSDP/TP executors.
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


def _ep_worker(rank: int, world: int, task, tid: int, batch_count: int,
               params: Optional[Dict[str, Any]], trial: bool):
    import torch
    import torch.distributed as dist

    from saturn_amd.executors.ddp import _make_optimizer
    from saturn_amd.parallel.ddp import BucketedDDP
    from saturn_amd.parallel.expert import (
        ep_expert_parameters,
        ep_full_state_dict,
        ep_scale_expert_grads,
        ep_shard_model,
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
        model = ep_shard_model(model)
        experts = ep_expert_parameters(model)
        # BucketedDDP broadcasts + all-reduces only the shared params;
        # expert shards stay rank-local by construction
        ddp = BucketedDDP(model, exclude=experts)
        optimizer = _make_optimizer(task, model)

        import os as _os

        opt_path = _os.path.join(
            task.save_dir, f"{task.name}.epopt.w{world}.r{rank}.pt"
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
            # DP over the batch dim: this rank's shard
            n = x.shape[0]
            lo = rank * n // world
            hi = (rank + 1) * n // world
            x = x[lo:hi].to(device, non_blocking=True)
            y = y[lo:hi].to(device, non_blocking=True)
            if x.is_floating_point():
                x = x.to(dtype)
            loss = task.loss_function(ddp(x), y)
            loss.backward()
            ddp.grad_sync()
            ep_scale_expert_grads(model)
            optimizer.step()
            optimizer.zero_grad(set_to_none=True)
            ddp.zero_grad_buffers()

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
        for _ in range(batch_count):
            step(next_batch())
        if device.type == "cuda":
            torch.cuda.synchronize()
        sd = ep_full_state_dict(model)
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


class ExpertParallelExecutor(BaseTechnique):
    """MoE expert parallelism + data parallelism over one RCCL gang."""

    name = "expert"

    @staticmethod
    def execute(task, gpus: List[int], tid: int, batch_count: int) -> None:
        params = (
            task.selected_strategy.parameters
            if task.selected_strategy is not None
            else {}
        )
        gang_spawn(_ep_worker, len(gpus), tid, task, tid, batch_count, params,
                   False)

    @staticmethod
    def search(
        task, gpus: List[int], tid: int
    ) -> Tuple[Optional[Dict[str, Any]], float]:
        if len(gpus) < 2:
            return None, float("inf")  # EP needs >1 rank to shard experts
        try:
            out = gang_spawn(
                _ep_worker, len(gpus), tid, task, tid, TRIAL_BATCHES,
                {"ep": len(gpus)}, True,
            )
        except Exception:
            return None, float("inf")
        if out is None:
            return None, float("inf")
        bt, hbm = out
        return dict(ep=len(gpus), hbm_peak_gb=round(hbm, 2)), bt
