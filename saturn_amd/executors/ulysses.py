"""Ulysses sequence-parallel executor (long-context training).

Beyond the reference's capability set (sequence parallelism is
absent there and named as the natural extension point — "it slots into the
Library as one more UDP ... the solver needs zero changes").  Contract:

- causal-LM tasks whose dataloader yields ``(tokens, tokens)`` batches
  (the shift is handled here, exactly, across shard boundaries);
- every rank sees the same batch and computes tokens
  ``[rank*T/P, (rank+1)*T/P)``; attention exchanges seq-shard for
  head-shard via all-to-all (saturn_amd.parallel.sequence) so the fused
  flash kernel gets the full sequence with exact causal masking;
- weight gradients are averaged with the bucketed DDP engine (ranks see
  different tokens), and the loss is scaled so the gradient equals the
  single-process full-sequence gradient exactly.

Requires T % world == 0 and (q and kv) head counts divisible by world.
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


def _sp_worker(rank: int, world: int, task, tid: int, batch_count: int,
               params: Optional[Dict[str, Any]], trial: bool):
    import torch
    import torch.distributed as dist

    from saturn_amd.executors.ddp import _make_optimizer
    from saturn_amd.ops.functional import fused_cross_entropy
    from saturn_amd.parallel.ddp import BucketedDDP
    from saturn_amd.parallel.sequence import sp_region

    backend = init_process_group(rank, world)
    try:
        device = (
            torch.device("cuda", rank) if backend == "nccl" else torch.device("cpu")
        )
        dtype = torch.bfloat16 if backend == "nccl" else torch.float32
        from saturn_amd.executors.ddp import build_model_on

        model = build_model_on(task, device, dtype)
        model.train()
        ddp = BucketedDDP(model, bucket_mb=float((params or {}).get("bucket_mb", 64.0)))
        optimizer = _make_optimizer(task, model)

        it = task.get_iterator() if not trial else task.get_fresh_iterator()

        def next_batch():
            nonlocal it
            try:
                return next(it)
            except StopIteration:
                it = task.get_fresh_iterator()
                return next(it)

        def step(batch):
            x, _y = batch
            x = x.to(device, non_blocking=True)
            B, T = x.shape
            assert T % world == 0, "seq length must divide the SP degree"
            Tl = T // world
            lo = rank * Tl
            x_loc = x[:, lo : lo + Tl].contiguous()
            # exact next-token labels for the local shard; the final
            # position of the LAST shard has no target
            labels = torch.full((B, Tl), -100, dtype=torch.long, device=device)
            hi = min(lo + Tl + 1, T)
            labels[:, : hi - lo - 1] = x[:, lo + 1 : hi]
            with sp_region(world, rank):
                logits = ddp(x_loc)
                local_mean = fused_cross_entropy(
                    logits, labels, shift=False, ignore_index=-100
                )
            # rescale so that after DDP's grad averaging the gradient
            # equals the global-mean-loss gradient exactly
            n_local = int((labels != -100).sum())
            n_total = B * (T - 1)
            loss = local_mean * (n_local * world / n_total)
            loss.backward()
            ddp.grad_sync()
            optimizer.step()
            ddp.zero_grad_buffers()
            return local_mean

        result = None
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
            result = ((timer() - t0) / (TRIAL_BATCHES - 1), hbm)
        else:
            last = None
            for _ in range(batch_count):
                last = step(next_batch())
            if device.type == "cuda":
                torch.cuda.synchronize()
            if rank == 0:
                import logging

                logging.getLogger(__name__).info(
                    "task %s (ulysses sp=%d): %d batches, local loss %.4f",
                    task.name, world, batch_count, float(last),
                )
                task.save_checkpoint(model, optimizer)
            if world > 1:
                dist.barrier()
        return result
    finally:
        destroy_process_group()


class UlyssesExecutor(BaseTechnique):
    """Sequence parallelism (Ulysses all-to-all) for long-context
    causal-LM tasks."""

    name = "ulysses"

    @staticmethod
    def execute(task, gpus: List[int], tid: int, batch_count: int) -> None:
        params = (
            task.selected_strategy.parameters
            if task.selected_strategy is not None
            else None
        )
        gang_spawn(_sp_worker, len(gpus), tid, task, tid, batch_count, params, False)

    @staticmethod
    def search(
        task, gpus: List[int], tid: int
    ) -> Tuple[Optional[Dict[str, Any]], float]:
        if len(gpus) < 2:
            return None, float("inf")  # SP over one GPU is plain training
        try:
            out = gang_spawn(
                _sp_worker, len(gpus), tid, task, tid, TRIAL_BATCHES,
                {"bucket_mb": 64.0}, True,
            )
        except Exception:
            return None, float("inf")
        if out is None:
            return None, float("inf")
        bt, hbm = out
        return {"sp": len(gpus), "bucket_mb": 64.0,
                "hbm_peak_gb": round(hbm, 2)}, bt
