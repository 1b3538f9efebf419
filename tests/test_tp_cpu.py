"""Tensor-parallel executor tests on CPU/gloo world_size=2."""

import pytest
import torch

from saturn_amd import HParams, Strategy, Task
from saturn_amd.executors.launch import gang_spawn
from saturn_amd.executors.megatron import MegatronExecutor
from saturn_amd.models.gptj import (
    get_gptj_model,
    make_token_dataloader,
    pretraining_loss,
)


def tiny_kwargs():
    return {"n_layer": 2, "n_embd": 64, "n_head": 2, "vocab_size": 128,
            "n_ctx": 32, "rotary_dim": 8}


def _tp2_worker(rank, world, _):
    import torch

    from saturn_amd.executors.launch import destroy_process_group, init_process_group
    from saturn_amd.parallel.tensor import tp_full_state_dict, tp_shard_model

    init_process_group(rank, world)
    try:
        torch.manual_seed(0)
        m = get_gptj_model(tiny_kwargs())
        ref = get_gptj_model(tiny_kwargs())  # identical init
        m = tp_shard_model(m)
        x = torch.randint(0, 128, (2, 32),
                          generator=torch.Generator().manual_seed(3))
        l_tp = pretraining_loss(m(x), x)
        l_ref = pretraining_loss(ref(x), x)
        assert abs(l_tp.item() - l_ref.item()) < 1e-4, (l_tp.item(), l_ref.item())
        l_tp.backward()
        # grads on replicated modules must match the unsharded model's
        l_ref.backward()
        g_tp = m.wte.weight.grad
        g_ref = ref.wte.weight.grad
        assert torch.allclose(g_tp, g_ref, atol=1e-5), "replicated grad mismatch"
        # sharded grads equal the dense model's corresponding slices:
        # column-parallel q_proj -> rows, row-parallel out_proj -> cols
        out_f = ref.h[0].attn.q_proj.weight.shape[0]
        sl = slice(rank * out_f // world, (rank + 1) * out_f // world)
        assert torch.allclose(
            m.h[0].attn.q_proj.weight.grad,
            ref.h[0].attn.q_proj.weight.grad[sl], atol=1e-5,
        ), "column-parallel shard grad mismatch"
        in_f = ref.h[0].attn.out_proj.weight.shape[1]
        slc = slice(rank * in_f // world, (rank + 1) * in_f // world)
        assert torch.allclose(
            m.h[0].attn.out_proj.weight.grad,
            ref.h[0].attn.out_proj.weight.grad[:, slc], atol=1e-5,
        ), "row-parallel shard grad mismatch"
        sd = tp_full_state_dict(m)
        if rank == 0:
            ref_sd = ref.state_dict()
            for k in ref_sd:
                assert torch.allclose(sd[k], ref_sd[k], atol=1e-6), k
            return True
        return None
    finally:
        destroy_process_group()


def test_tp2_forward_matches_and_state_dict_roundtrip():
    assert gang_spawn(_tp2_worker, 2, 930, None, timeout=300) is True


def test_megatron_executor_search_and_execute(save_dir):
    t = Task(
        lambda kwargs=None: get_gptj_model(tiny_kwargs()),
        make_token_dataloader(batch_size=2, seq_len=32, vocab=128, n_batches=8),
        pretraining_loss,
        HParams(lr=1e-3, batch_count=4),
        name="tp_t",
        save_dir=save_dir,
    )
    params, bt = MegatronExecutor.search(t, [0, 1], 931)
    assert params is not None and params.get("tp") == 2 and bt > 0
    t.strategies[2] = Strategy(MegatronExecutor, 2, params, bt * 4, batch_time=bt)
    t.select_strategy(t.strategies[2])
    MegatronExecutor.execute(t, [0, 1], 931, 2)
    assert t.has_ckpt()
    # checkpoint must load into a fresh unsharded model
    m = t.get_model()
    assert m is not None


def test_megatron_rejects_single_gpu(save_dir):
    t = Task(
        lambda kwargs=None: get_gptj_model(tiny_kwargs()),
        make_token_dataloader(batch_size=2, seq_len=32, vocab=128, n_batches=8),
        pretraining_loss,
        HParams(lr=1e-3, batch_count=4),
        name="tp_1g",
        save_dir=save_dir,
    )
    params, _ = MegatronExecutor.search(t, [0], 932)
    assert params is None


def test_tp_optimizer_shard_resume(save_dir, library_path):
    """Second interval reloads each rank's optimizer shard (Adam moments)."""
    import os

    import torch

    from saturn_amd import HParams, Strategy, Task
    from saturn_amd.executors.megatron import MegatronExecutor
    from saturn_amd.models import get_mlp_dataloader, get_mlp_model, mse_loss

    t = Task(
        get_mlp_model,
        get_mlp_dataloader,
        mse_loss,
        HParams(lr=1e-2, batch_count=6, optimizer_cls=torch.optim.Adam),
        gpu_range=[2],
        name="tp_opt",
        save_dir=save_dir,
    )
    t.strategies[2] = Strategy(MegatronExecutor, 2, {"tp": 2}, 6.0,
                               batch_time=1.0)
    t.select_strategy(t.strategies[2])
    MegatronExecutor.execute(t, [0, 1], 931, 3)
    for r in (0, 1):
        shard = os.path.join(save_dir, f"tp_opt.tpopt.w2.r{r}.pt")
        assert os.path.isfile(shard), shard
        st = torch.load(shard, weights_only=False)
        assert any("exp_avg" in v for v in st["state"].values())
    MegatronExecutor.execute(t, [0, 1], 931, 3)  # second interval loads back
    assert t.has_ckpt()
