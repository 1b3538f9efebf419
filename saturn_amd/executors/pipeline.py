"""GPipe pipeline executor (single process, multi-GPU over xGMI).

Capability parity with the reference Pipeline UDP
(``examples/wikitext103/executors/Pipeline.py:24-167``): requires a model
flattenable to ``nn.Sequential`` (hint ``to_sequential`` or a known model
type), autotunes the microbatch count with the reference's halving search
(Pipeline.py:139-159) and balances stages; runs on
``saturn_amd.parallel.pipeline.PipelinedModel`` instead of torchgpipe.
"""

from __future__ import annotations

from timeit import default_timer as timer
from typing import Any, Dict, List, Optional, Tuple

from saturn_amd.core.technique import BaseTechnique

TRIAL_BATCHES = 3


def _to_sequential(task, model):
    import torch.nn as nn

    fn = (task.hints or {}).get("to_sequential")
    if fn is not None:
        return fn(model)
    if isinstance(model, nn.Sequential):
        return model
    from saturn_amd.models.gptj import GPTJForCausalLM
    from saturn_amd.models.gptj import as_sequential as gptj_seq

    if isinstance(model, GPTJForCausalLM):
        return gptj_seq(model)
    from saturn_amd.models.gpt2 import GPT2ForCausalLM
    from saturn_amd.models.gpt2 import as_sequential as gpt2_seq

    if isinstance(model, GPT2ForCausalLM):
        return gpt2_seq(model)
    from saturn_amd.models.llama import LlamaForCausalLM
    from saturn_amd.models.llama import as_sequential as llama_seq

    if isinstance(model, LlamaForCausalLM):
        return llama_seq(model)
    raise ValueError(
        "Pipeline executor needs an nn.Sequential model or a "
        "hints['to_sequential'] flattener."
    )


def _run_pipeline(task, n_devices: int, batch_count: int,
                  params: Optional[Dict[str, Any]], trial: bool):
    import torch

    from saturn_amd.executors.ddp import _make_optimizer
    from saturn_amd.parallel.pipeline import PipelinedModel

    params = params or {}
    use_gpu = torch.cuda.is_available()
    devices = (
        [torch.device("cuda", i) for i in range(n_devices)]
        if use_gpu
        else [torch.device("cpu")] * n_devices
    )
    dtype = torch.bfloat16 if use_gpu else torch.float32

    from saturn_amd.executors.ddp import build_model_on

    model = build_model_on(task, devices[0], dtype)
    seq = _to_sequential(task, model)
    balance = params.get("balance")
    if balance is None and params.get("time_balance", True):
        # profile per-layer time on device 0 (reference balance_by_time,
        # Pipeline.py:94-103); fall back to parameter balance if the whole
        # model cannot visit one device
        try:
            from saturn_amd.parallel.pipeline import balance_by_time

            x0, _ = next(task.get_fresh_iterator())
            sample = x0[: max(1, x0.shape[0] // 4)]
            if sample.is_floating_point():
                sample = sample.to(dtype)
            balance = balance_by_time(seq, sample, n_devices, devices[0])
        except Exception:
            balance = None
    pipe = PipelinedModel(
        seq,
        devices,
        balance=balance,
        chunks=int(params.get("chunks", 4)),
        checkpoint_activations=bool(params.get("checkpoint", False)),
    )
    pipe.train()
    optimizer = _make_optimizer(task, pipe)

    it = task.get_iterator() if not trial else task.get_fresh_iterator()

    def next_batch():
        nonlocal it
        try:
            return next(it)
        except StopIteration:
            it = task.get_fresh_iterator()
            return next(it)

    last_dev = devices[-1]

    def step(batch):
        x, y = batch
        if x.is_floating_point():
            x = x.to(dtype)
        out = pipe(x)
        loss = task.loss_function(out, y.to(last_dev, non_blocking=True))
        loss.backward()
        optimizer.step()
        optimizer.zero_grad(set_to_none=True)

    def sync():
        if use_gpu:
            for d in devices:
                torch.cuda.synchronize(d)

    if trial:
        step(next_batch())
        sync()
        if use_gpu:
            for d in devices:
                torch.cuda.reset_peak_memory_stats(d)
        t0 = timer()
        for _ in range(TRIAL_BATCHES - 1):
            step(next_batch())
        sync()
        hbm = (
            max(torch.cuda.max_memory_allocated(d) for d in devices) / 2**30
            if use_gpu
            else 0.0
        )
        return ((timer() - t0) / (TRIAL_BATCHES - 1), hbm)
    for _ in range(batch_count):
        step(next_batch())
    sync()
    task.save_checkpoint(model, optimizer)
    return None


class PipelineExecutor(BaseTechnique):
    """GPipe microbatch pipelining over xGMI."""

    name = "pipeline"

    @staticmethod
    def execute(task, gpus: List[int], tid: int, batch_count: int) -> None:
        params = (
            task.selected_strategy.parameters
            if task.selected_strategy is not None
            else {}
        )
        _run_pipeline(task, len(gpus), batch_count, params, False)

    @staticmethod
    def search(
        task, gpus: List[int], tid: int
    ) -> Tuple[Optional[Dict[str, Any]], float]:
        if len(gpus) < 2:
            return None, float("inf")  # pipelining needs >= 2 stages
        # probe the batch size to bound the chunk search
        try:
            first = next(task.get_fresh_iterator())
            bsz = first[0].shape[0]
        except Exception:
            bsz = 8
        chunks = 1
        grid = []
        while chunks <= min(8, bsz):
            grid.append(chunks)
            chunks *= 2
        best: Tuple[Optional[Dict[str, Any]], float] = (None, float("inf"))
        for c in reversed(grid):  # most chunks first (reference halves down)
            # activation-checkpoint knob: plain first, ckpt as the
            # memory-rescue variant (first-fit, like the FSDP grid)
            for ckpt in (False, True):
                try:
                    bt, hbm = _run_pipeline(
                        task, len(gpus), 0,
                        {"chunks": c, "checkpoint": ckpt}, True,
                    )
                except Exception:
                    continue
                if bt < best[1]:
                    best = ({"chunks": c, "checkpoint": ckpt,
                             "hbm_peak_gb": round(hbm, 2)}, bt)
                break  # plain run fit -> no need to pay recompute
        return best
