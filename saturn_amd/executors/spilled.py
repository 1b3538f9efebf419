"""Spilled executor: single-GPU training of models larger than HBM by
streaming layer groups between pinned host DRAM and the GPU.

Capability parity with the reference Spilled UDP
(``examples/wikitext103/executors/Spilled.py:23-152``, fairscale
OffloadModel with checkpoint_activation=True): partitions the block list
into k groups (k searched over divisors of the layer count,
Spilled.py:91-107) and runs the framework's shard engine with
world_size=1 + host offload — in that configuration
``saturn_amd.parallel.zero3``'s gather/free cycle IS double-buffered
host<->HBM layer streaming, with activation checkpointing per
group and the optimizer stepping on the host-resident flat shards.

Sized for MI355X: with 288 GB HBM spilling only pays for >=70B-class
models, so ``search`` first checks whether the model fits outright and
reports infeasible only on real OOM.
"""

from __future__ import annotations

from timeit import default_timer as timer
from typing import Any, Dict, List, Optional, Tuple

from saturn_amd.core.technique import BaseTechnique

TRIAL_BATCHES = 3


def _blocks_of(task, model):
    fn = (task.hints or {}).get("get_blocks")
    if fn is not None:
        return list(fn(model))
    if hasattr(model, "h"):
        return list(model.h)
    import torch.nn as nn

    if isinstance(model, nn.Sequential):
        return list(model)
    raise ValueError(
        "Spilled executor needs model.h, an nn.Sequential, or a "
        "hints['get_blocks'] accessor."
    )


def _group(blocks, n_partitions: int):
    """Split blocks into n_partitions contiguous groups."""
    import torch.nn as nn

    per = (len(blocks) + n_partitions - 1) // n_partitions
    return [
        nn.ModuleList(blocks[i : i + per]) for i in range(0, len(blocks), per)
    ]


def _run_spilled(task, batch_count: int, params: Optional[Dict[str, Any]],
                 trial: bool):
    import torch

    from saturn_amd.executors.ddp import _make_optimizer
    from saturn_amd.parallel.zero3 import Zero3Model

    params = params or {}
    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda", 0) if use_gpu else torch.device("cpu")
    dtype = torch.bfloat16 if use_gpu else torch.float32

    from saturn_amd.executors.ddp import build_model_on

    model = build_model_on(task, device, dtype)
    model.train()
    blocks = _blocks_of(task, model)
    n_part = int(params.get("partitions", len(blocks)))
    units = _group(blocks, n_part)
    z3 = Zero3Model(
        model,
        unit_modules=units,
        device=device,
        offload=bool(params.get("offload", use_gpu)),
        checkpoint_activations=True,
        prefetch=True,
    )

    import torch.nn as nn

    class _Holder(nn.Module):
        def __init__(self, shards):
            super().__init__()
            self.ps = nn.ParameterList(shards)

    optimizer = _make_optimizer(task, _Holder(z3.sharded_parameters()))
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
        loss = task.loss_function(z3(x), y)
        loss.backward()
        z3.grad_sync()
        optimizer.step()
        z3.zero_grad_shards()

    if trial:
        step(next_batch())
        if use_gpu:
            torch.cuda.synchronize()
            torch.cuda.reset_peak_memory_stats()
        t0 = timer()
        for _ in range(TRIAL_BATCHES - 1):
            step(next_batch())
        if use_gpu:
            torch.cuda.synchronize()
        hbm = (
            torch.cuda.max_memory_allocated() / 2**30 if use_gpu else 0.0
        )
        return ((timer() - t0) / (TRIAL_BATCHES - 1), hbm)
    for _ in range(batch_count):
        step(next_batch())
    if use_gpu:
        torch.cuda.synchronize()
    sd = z3.full_state_dict()
    if sd is not None:
        task.save_checkpoint(sd, None)
    return None


class SpilledExecutor(BaseTechnique):
    """Host-DRAM spilling for models beyond HBM (single GPU,
    reference Spilled.py:27-28 is also single-GPU-only)."""

    name = "spilled"

    @staticmethod
    def execute(task, gpus: List[int], tid: int, batch_count: int) -> None:
        params = (
            task.selected_strategy.parameters
            if task.selected_strategy is not None
            else {}
        )
        _run_spilled(task, batch_count, params, False)

    @staticmethod
    def search(
        task, gpus: List[int], tid: int
    ) -> Tuple[Optional[Dict[str, Any]], float]:
        if len(gpus) != 1:
            return None, float("inf")
        # Spilling exists for models that DON'T fit HBM: layer streaming is
        # strictly slower than resident training when everything fits, so
        # running the divisor-grid trials for a comfortably-fitting model
        # only burns search time (round-2 config-4: 290 s of search for a
        # 48 s makespan, mostly spill trials the solver never picked).
        # A meta-device build counts parameters for free.
        try:
            import torch

            if torch.cuda.is_available():
                with torch.device("meta"):
                    meta_model = task.get_model(fresh=True)
                pbytes = sum(
                    p.numel() * 2 for p in meta_model.parameters()  # bf16
                )
                del meta_model
                _free, total = torch.cuda.mem_get_info()
                # weights + grads + optimizer + activation slack ~4x
                if pbytes * 4 < total * 0.7:
                    return None, float("inf")  # resident techniques win
        except Exception:
            pass  # meta build unsupported -> fall through to real trials
        # probe block count for the divisor grid (reference Spilled.py:91-96)
        try:
            model = task.get_model(fresh=True)
            n_blocks = len(_blocks_of(task, model))
            del model
        except Exception:
            return None, float("inf")
        divisors = [d for d in range(1, n_blocks + 1) if n_blocks % d == 0]
        # few partitions = fewer, larger transfers; try coarsest first
        best: Tuple[Optional[Dict[str, Any]], float] = (None, float("inf"))
        for d in divisors[:4]:
            try:
                bt, hbm = _run_spilled(task, 0, {"partitions": d}, True)
            except Exception:
                continue
            if bt < best[1]:
                best = ({"partitions": d, "hbm_peak_gb": round(hbm, 2)}, bt)
            if best[0] is not None and d > 1:
                break  # first fitting coarse partition wins (transfer-bound)
        return best
