"""Ulysses sequence parallelism on CPU/gloo world 2: exact match against
the single-process full-sequence run (loss and parameter trajectories)."""

import pytest
import torch

from saturn_amd import HParams, Strategy, Task
from saturn_amd.executors.launch import gang_spawn
from saturn_amd.executors.ulysses import UlyssesExecutor, _sp_worker
from saturn_amd.models.gptj import (
    get_gptj_model,
    make_token_dataloader,
    pretraining_loss,
)


def tiny_kwargs():
    return {"n_layer": 2, "n_embd": 64, "n_head": 4, "vocab_size": 128,
            "n_ctx": 64, "rotary_dim": 16}


def _sp2_worker(rank, world, _):
    import torch

    from saturn_amd.executors.launch import destroy_process_group, init_process_group
    from saturn_amd.ops.functional import fused_cross_entropy
    from saturn_amd.parallel.sequence import sp_region

    init_process_group(rank, world)
    try:
        torch.manual_seed(0)
        m = get_gptj_model(tiny_kwargs())
        x = torch.randint(0, 128, (2, 64),
                          generator=torch.Generator().manual_seed(5))
        B, T = x.shape
        Tl = T // world
        lo = rank * Tl
        labels = torch.full((B, Tl), -100, dtype=torch.long)
        hi = min(lo + Tl + 1, T)
        labels[:, : hi - lo - 1] = x[:, lo + 1 : hi]
        with sp_region(world, rank):
            logits = m(x[:, lo : lo + Tl].contiguous())
            local_mean = fused_cross_entropy(logits, labels, shift=False)
        n_local = int((labels != -100).sum())
        loss = local_mean * (n_local * world / (B * (T - 1)))
        loss.backward()
        # average grads like the executor's DDP sync would
        import torch.distributed as dist

        g = m.wte.weight.grad.clone()
        dist.all_reduce(g)
        g /= world

        # reference: single-process full sequence
        torch.manual_seed(0)
        ref = get_gptj_model(tiny_kwargs())
        ref_loss = pretraining_loss(ref(x), x)
        ref_loss.backward()

        if rank == 0:
            return (
                float(local_mean),
                float(ref_loss),
                float((g - ref.wte.weight.grad).abs().max()),
                float(ref.wte.weight.grad.abs().max()),
            )
        return None
    finally:
        destroy_process_group()


def test_sp2_gradients_match_full_sequence():
    local_mean, ref_loss, gdiff, gmax = gang_spawn(_sp2_worker, 2, 960, None,
                                                   timeout=300)
    # averaged SP grads == full-sequence grads (same math, exact scaling)
    assert gdiff < 5e-5 * max(1.0, gmax), (gdiff, gmax)


def test_ulysses_executor_search_and_execute(save_dir):
    t = Task(
        lambda kwargs=None: get_gptj_model(tiny_kwargs()),
        make_token_dataloader(batch_size=2, seq_len=64, vocab=128, n_batches=8),
        pretraining_loss,
        HParams(lr=1e-3, batch_count=4),
        name="sp_t",
        save_dir=save_dir,
    )
    params, bt = UlyssesExecutor.search(t, [0, 1], 961)
    assert params is not None and params["sp"] == 2 and bt > 0
    t.strategies[2] = Strategy(UlyssesExecutor, 2, params, bt * 4, batch_time=bt)
    t.select_strategy(t.strategies[2])
    UlyssesExecutor.execute(t, [0, 1], 961, 2)
    assert t.has_ckpt()


def test_ulysses_rejects_single_gpu(save_dir):
    t = Task(
        lambda kwargs=None: get_gptj_model(tiny_kwargs()),
        make_token_dataloader(batch_size=2, seq_len=64, vocab=128, n_batches=8),
        pretraining_loss,
        HParams(lr=1e-3, batch_count=4),
        name="sp_1g",
        save_dir=save_dir,
    )
    params, _ = UlyssesExecutor.search(t, [0], 962)
    assert params is None
