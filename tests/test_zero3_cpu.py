"""ZeRO-3 shard manager tests on CPU (world 1 in-process, world 2 gloo)."""

import pytest
import torch

from saturn_amd import HParams, Task
from saturn_amd.executors.fsdp import FSDPExecutor, _fsdp_worker
from saturn_amd.executors.launch import gang_spawn
from saturn_amd.models import get_mlp_dataloader, get_mlp_model, mse_loss
from saturn_amd.models.gptj import get_gptj_model, pretraining_loss


def tiny_gptj():
    return get_gptj_model(
        {"n_layer": 2, "n_embd": 64, "n_head": 2, "vocab_size": 128,
         "n_ctx": 32, "rotary_dim": 8}
    )


def test_zero3_world1_matches_plain_training():
    """world=1 sharded training must match unsharded step-for-step."""
    from saturn_amd.parallel.zero3 import Zero3Model

    torch.manual_seed(0)
    m1 = tiny_gptj()
    m2 = tiny_gptj()  # same seed -> same init
    for a, b in zip(m1.parameters(), m2.parameters()):
        assert torch.equal(a, b)

    x = torch.randint(0, 128, (2, 32))
    z3 = Zero3Model(m2, prefetch=False)
    opt1 = torch.optim.SGD(m1.parameters(), lr=0.1)
    opt2 = torch.optim.SGD(z3.sharded_parameters(), lr=0.1)
    for _ in range(3):
        l1 = pretraining_loss(m1(x), x)
        l1.backward()
        opt1.step()
        m1.zero_grad()

        l2 = pretraining_loss(z3(x), x)
        l2.backward()
        z3.grad_sync()
        opt2.step()
        z3.zero_grad_shards()
        assert abs(l1.item() - l2.item()) < 1e-4, (l1.item(), l2.item())


def test_zero3_checkpoint_activations_world1():
    from saturn_amd.parallel.zero3 import Zero3Model

    torch.manual_seed(0)
    m = tiny_gptj()
    ref = tiny_gptj()
    x = torch.randint(0, 128, (2, 32))
    z3 = Zero3Model(m, prefetch=False, checkpoint_activations=True)
    l_ref = pretraining_loss(ref(x), x)
    l = pretraining_loss(z3(x), x)
    assert abs(l.item() - l_ref.item()) < 1e-4
    l.backward()
    z3.grad_sync()
    assert all(u.shard.grad is not None for u in z3.units)


def test_zero3_full_state_dict_roundtrip():
    from saturn_amd.parallel.zero3 import Zero3Model

    torch.manual_seed(0)
    m = tiny_gptj()
    ref_sd = {k: v.clone() for k, v in m.state_dict().items()}
    z3 = Zero3Model(m, prefetch=False)
    sd = z3.full_state_dict()
    for k in ref_sd:
        assert torch.equal(sd[k], ref_sd[k]), k


def _z3_world2_worker(rank, world, state):
    import torch

    from saturn_amd.executors.launch import destroy_process_group, init_process_group
    from saturn_amd.parallel.zero3 import Zero3Model

    init_process_group(rank, world)
    try:
        torch.manual_seed(rank)  # deliberately different init; bcast fixes it
        m = tiny_gptj()
        z3 = Zero3Model(m, prefetch=False)
        x = torch.randint(0, 128, (2, 32), generator=torch.Generator().manual_seed(7))
        opt = torch.optim.SGD(z3.sharded_parameters(), lr=0.05)
        losses = []
        for _ in range(3):
            from saturn_amd.models.gptj import pretraining_loss

            loss = pretraining_loss(z3(x), x)
            loss.backward()
            z3.grad_sync()
            opt.step()
            z3.zero_grad_shards()
            losses.append(float(loss))
        sd = z3.full_state_dict()
        if rank == 0:
            return losses, {k: v for k, v in sd.items()}
        return None
    finally:
        destroy_process_group()


def test_zero3_world2_gloo_matches_world1():
    """Same data on both ranks -> grads identical to single-process run."""
    out = gang_spawn(_z3_world2_worker, 2, 910, None, timeout=300)
    losses2, sd2 = out

    torch.manual_seed(0)
    from saturn_amd.parallel.zero3 import Zero3Model

    m = tiny_gptj()
    z3 = Zero3Model(m, prefetch=False)
    x = torch.randint(0, 128, (2, 32), generator=torch.Generator().manual_seed(7))
    opt = torch.optim.SGD(z3.sharded_parameters(), lr=0.05)
    losses1 = []
    for _ in range(3):
        loss = pretraining_loss(z3(x), x)
        loss.backward()
        z3.grad_sync()
        opt.step()
        z3.zero_grad_shards()
        losses1.append(float(loss))
    for a, b in zip(losses1, losses2):
        assert abs(a - b) < 1e-4, (losses1, losses2)


def test_fsdp_executor_search_and_execute(save_dir, library_path):
    t = Task(
        get_mlp_model,
        get_mlp_dataloader,
        mse_loss,
        HParams(lr=1e-2, batch_count=4),
        gpu_range=[1, 2],
        name="fsdp_mlp",
        save_dir=save_dir,
    )
    params, bt = FSDPExecutor.search(t, [0, 1], 911)
    assert params is not None and bt > 0
    t.strategies[2] = __import__("saturn_amd").Strategy(
        FSDPExecutor, 2, params, bt * 4, batch_time=bt
    )
    t.select_strategy(t.strategies[2])
    FSDPExecutor.execute(t, [0, 1], 911, 2)
    assert t.has_ckpt()


def test_fsdp_optimizer_shard_resume(save_dir, library_path):
    """Second interval resumes optimizer shard state (same world size)."""
    import os

    t = Task(
        get_mlp_model,
        get_mlp_dataloader,
        mse_loss,
        HParams(lr=1e-2, batch_count=6,
                optimizer_cls=__import__("torch").optim.Adam),
        gpu_range=[1],
        name="fsdp_opt",
        save_dir=save_dir,
    )
    params, bt = FSDPExecutor.search(t, [0], 915)
    assert params is not None
    t.strategies[1] = __import__("saturn_amd").Strategy(
        FSDPExecutor, 1, params, bt * 6, batch_time=bt
    )
    t.select_strategy(t.strategies[1])
    FSDPExecutor.execute(t, [0], 915, 3)
    shard_file = os.path.join(save_dir, "fsdp_opt.optshard.w1.r0.pt")
    assert os.path.isfile(shard_file)
    import torch

    st = torch.load(shard_file, weights_only=False)
    assert any("exp_avg" in v for v in st["state"].values())
    FSDPExecutor.execute(t, [0], 915, 3)  # second interval loads it back
    assert t.has_ckpt()
