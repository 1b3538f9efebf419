"""Expert-parallel tests on CPU (world 1 in-process, world 2 gloo).

The world-2 test pins EP's gradient semantics exactly against
single-process dense training on the same global batch: shared params via
DDP-mean, expert grads via the 1/world rescale (parallel/expert.py
docstring)."""

import torch

from saturn_amd.executors.launch import gang_spawn
from saturn_amd.models.mixtral import get_mixtral_model, mixtral_loss

TINY = {"n_layer": 2, "n_embd": 64, "n_head": 2, "n_kv_head": 1,
        "vocab_size": 128, "n_ctx": 32, "ffn_dim": 96, "n_expert": 4,
        "top_k": 2}


def test_moe_dense_forward_backward():
    torch.manual_seed(0)
    m = get_mixtral_model(TINY)
    x = torch.randint(0, 128, (2, 32))
    loss = mixtral_loss(m(x), x)
    loss.backward()
    assert torch.isfinite(loss.detach())
    # router grads flow (gate weights multiply expert outputs)
    assert m.h[0].mlp.router.weight.grad is not None
    assert m.h[0].mlp.router.weight.grad.abs().sum() > 0


def test_ep_world1_matches_dense():
    from saturn_amd.parallel.expert import ep_shard_model

    torch.manual_seed(0)
    m1 = get_mixtral_model(TINY)
    m2 = ep_shard_model(get_mixtral_model(TINY))
    x = torch.randint(0, 128, (2, 32))
    l1 = mixtral_loss(m1(x), x)
    l2 = mixtral_loss(m2(x), x)
    assert abs(float(l1) - float(l2)) < 1e-5


def _ep_world2_worker(rank, world, state):
    import torch
    import torch.distributed as dist

    from saturn_amd.executors.launch import (
        destroy_process_group,
        init_process_group,
    )
    from saturn_amd.models.mixtral import get_mixtral_model, mixtral_loss
    from saturn_amd.parallel.ddp import BucketedDDP
    from saturn_amd.parallel.expert import (
        ep_expert_parameters,
        ep_full_state_dict,
        ep_scale_expert_grads,
        ep_shard_model,
    )

    TINY = {"n_layer": 2, "n_embd": 64, "n_head": 2, "n_kv_head": 1,
            "vocab_size": 128, "n_ctx": 32, "ffn_dim": 96, "n_expert": 4,
            "top_k": 2}
    init_process_group(rank, world)
    try:
        torch.manual_seed(0)
        model = ep_shard_model(get_mixtral_model(TINY))
        experts = ep_expert_parameters(model)
        ddp = BucketedDDP(model, exclude=experts)
        gx = torch.randint(0, 128, (4, 32),
                           generator=torch.Generator().manual_seed(11))
        x = gx[rank * 2:(rank + 1) * 2]
        loss = mixtral_loss(ddp(x), x)
        loss.backward()
        ddp.grad_sync()
        ep_scale_expert_grads(model)

        sd = ep_full_state_dict(model)
        if rank == 0:
            owned = {}
            for li in range(2):
                moe = model.h[li].mlp
                for i, e in enumerate(moe.local_experts):
                    owned[(li, moe.e0 + i)] = {
                        "gate": e.gate_proj.weight.grad.clone(),
                        "down": e.down_proj.weight.grad.clone(),
                    }
            return (
                float(loss),
                model.h[0].mlp.router.weight.grad.clone(),
                model.wte.weight.grad.clone(),
                owned,
                sd,
            )
        return None
    finally:
        destroy_process_group()


def test_ep_world2_matches_single_process():
    out = gang_spawn(_ep_world2_worker, 2, 950, None, timeout=300)
    _, router_g, wte_g, owned, sd = out

    # single-process dense reference on the SAME global batch
    torch.manual_seed(0)
    ref = get_mixtral_model(TINY)
    gx = torch.randint(0, 128, (4, 32),
                       generator=torch.Generator().manual_seed(11))
    loss = mixtral_loss(ref(gx), gx)
    loss.backward()

    assert torch.allclose(ref.h[0].mlp.router.weight.grad, router_g,
                          atol=1e-5), "router (shared) grad mismatch"
    assert torch.allclose(ref.wte.weight.grad, wte_g, atol=1e-5)
    for (li, g), grads in owned.items():
        e = ref.h[li].mlp.experts[g]
        assert torch.allclose(e.gate_proj.weight.grad, grads["gate"],
                              atol=1e-5), (li, g)
        assert torch.allclose(e.down_proj.weight.grad, grads["down"],
                              atol=1e-5), (li, g)

    # reassembled full state dict matches the dense module tree exactly
    ref_keys = set(ref.state_dict().keys())
    sd_keys = set(sd.keys())
    missing = {k for k in ref_keys - sd_keys if "rope" not in k}
    assert not missing, missing
    for k in sd:
        if k in ref_keys:
            assert sd[k].shape == ref.state_dict()[k].shape, k


def test_ep_executor_search_and_execute(save_dir, library_path):
    """Full technique-library path: EP trials on 2 gloo ranks, then an
    interval execution with checkpoint + optimizer shards."""
    import os

    from saturn_amd import HParams, Strategy, Task
    from saturn_amd.executors.expert import ExpertParallelExecutor
    from saturn_amd.models.gptj import make_token_dataloader

    t = Task(
        lambda kwargs=None: get_mixtral_model(TINY),
        make_token_dataloader(batch_size=4, seq_len=32, vocab=128,
                              n_batches=6),
        mixtral_loss,
        HParams(lr=1e-2, batch_count=4,
                optimizer_cls=torch.optim.Adam),
        gpu_range=[2],
        name="ep_moe",
        save_dir=save_dir,
    )
    params, bt = ExpertParallelExecutor.search(t, [0, 1], 955)
    assert params is not None and params.get("ep") == 2 and bt > 0
    t.strategies[2] = Strategy(ExpertParallelExecutor, 2, params, bt * 4,
                               batch_time=bt)
    t.select_strategy(t.strategies[2])
    ExpertParallelExecutor.execute(t, [0, 1], 955, 2)
    assert t.has_ckpt()
    for r in (0, 1):
        assert os.path.isfile(
            os.path.join(save_dir, f"ep_moe.epopt.w2.r{r}.pt"))
    # checkpoint is the DENSE layout: a fresh dense model can load it
    dense = get_mixtral_model(TINY)
    ck = torch.load(t.ckpt_path, weights_only=False)
    sd = ck["model"] if isinstance(ck, dict) and "model" in ck else ck
    missing, unexpected = dense.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ExpertParallelExecutor.execute(t, [0, 1], 955, 2)  # resume interval
