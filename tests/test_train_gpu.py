"""GPU integration: tiny GPT-J trains (loss decreases) on the fused stack,
and the DDP executor's single-GPU path works end to end."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_gptj_tiny_loss_decreases():
    from saturn_amd.models.gptj import GPTJConfig, GPTJForCausalLM, pretraining_loss
    from saturn_amd.ops import require_ext
    from saturn_amd.ops.optim import FusedAdam

    require_ext()
    torch.manual_seed(0)
    cfg = GPTJConfig(n_layer=2, n_embd=512, n_head=4, n_ctx=128,
                     vocab_size=2048, rotary_dim=32)
    model = GPTJForCausalLM(cfg).to("cuda", torch.bfloat16)
    opt = FusedAdam(model.parameters(), lr=3e-4)
    x = torch.randint(0, cfg.vocab_size, (4, 128), device="cuda")
    losses = []
    for _ in range(20):
        loss = pretraining_loss(model(x), x)
        loss.backward()
        opt.step()
        for p in model.parameters():
            p.grad = None
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.7, f"no learning: {losses[0]} -> {losses[-1]}"


def test_bucketed_ddp_single_gpu_step():
    from saturn_amd.models.gptj import GPTJConfig, GPTJForCausalLM, pretraining_loss
    from saturn_amd.ops.optim import FusedSGD
    from saturn_amd.parallel.ddp import BucketedDDP

    torch.manual_seed(0)
    cfg = GPTJConfig(n_layer=2, n_embd=256, n_head=4, n_ctx=64,
                     vocab_size=1024, rotary_dim=16)
    model = GPTJForCausalLM(cfg).to("cuda", torch.bfloat16)
    ddp = BucketedDDP(model, bucket_mb=4)
    opt = FusedSGD(model.parameters(), lr=1e-3)
    x = torch.randint(0, cfg.vocab_size, (2, 64), device="cuda")
    for _ in range(2):
        loss = pretraining_loss(ddp(x), x)
        loss.backward()
        ddp.grad_sync()
        opt.step()
        ddp.zero_grad_buffers()
    assert torch.isfinite(torch.tensor(float(loss)))


def test_native_extension_is_loaded():
    """Round-end check mirror: the in-tree .so must be the loaded one."""
    import saturn_amd._C as C

    assert "/saturn_amd/" in C.__file__, C.__file__
    assert C.__file__.endswith(".so")


def test_native_comm_world1_roundtrip():
    """Native RCCL engine: world-1 communicator, all_reduce/broadcast/
    all_gather/reduce_scatter smoke + BucketedDDP on the native engine."""
    from saturn_amd.comm import has_native_comm, require_native_comm

    if not has_native_comm():
        import pytest

        pytest.fail("native comm extension not built")
    ext = require_native_comm()
    comm = ext.RcclComm(ext.get_unique_id(), 0, 1)
    x = torch.randn(1024, device="cuda", dtype=torch.bfloat16)
    ref = x.clone()
    comm.all_reduce(x, True)
    comm.join()
    torch.cuda.synchronize()
    assert torch.equal(x, ref)  # world 1: identity
    full = torch.empty(1024, device="cuda", dtype=torch.bfloat16)
    comm.all_gather(full, x)
    comm.join()
    torch.cuda.synchronize()
    assert torch.equal(full, ref)

    from saturn_amd.models.mlp import get_mlp_model, mse_loss
    from saturn_amd.parallel.ddp import BucketedDDP

    m = get_mlp_model().to("cuda", torch.bfloat16)
    ddp = BucketedDDP(m, bucket_mb=1.0, comm=comm)
    xx = torch.randn(8, 32, device="cuda", dtype=torch.bfloat16)
    yy = torch.randn(8, 8, device="cuda", dtype=torch.bfloat16)
    loss = mse_loss(ddp(xx), yy)
    loss.backward()
    ddp.grad_sync()
    assert all(torch.isfinite(b.flat).all() for b in ddp.buckets)


def test_zero3_offload_spill_path_gpu():
    """World-1 ZeRO-3 with host offload = the spill engine on hardware:
    pinned shard -> HBM gather per unit, activation ckpt, CPU optimizer."""
    from saturn_amd.models.gptj import GPTJConfig, GPTJForCausalLM, pretraining_loss
    from saturn_amd.parallel.zero3 import Zero3Model

    torch.manual_seed(0)
    cfg = GPTJConfig(n_layer=4, n_embd=512, n_head=4, n_ctx=128,
                     vocab_size=2048, rotary_dim=32)
    m = GPTJForCausalLM(cfg).to("cuda", torch.bfloat16)
    z3 = Zero3Model(m, offload=True, checkpoint_activations=True)
    for u in z3.units:
        assert u.shard.device.type == "cpu"  # spilled to host
    opt = torch.optim.SGD(z3.sharded_parameters(), lr=1e-3)
    x = torch.randint(0, cfg.vocab_size, (2, 128), device="cuda")
    for _ in range(2):
        loss = pretraining_loss(z3(x), x)
        loss.backward()
        z3.grad_sync()
        opt.step()
        z3.zero_grad_shards()
    assert torch.isfinite(torch.tensor(float(loss)))
    sd = z3.full_state_dict()
    assert all(v.device.type == "cpu" for v in sd.values())


def test_fsdp_executor_world1_gpu(tmp_path):
    from saturn_amd import HParams, Task
    from saturn_amd.executors.fsdp import FSDPExecutor
    from saturn_amd.models.gptj import get_gptj_model, make_token_dataloader, pretraining_loss

    t = Task(
        lambda kwargs=None: get_gptj_model(
            {"n_layer": 2, "n_embd": 256, "n_head": 4, "vocab_size": 512,
             "n_ctx": 64, "rotary_dim": 16}
        ),
        make_token_dataloader(batch_size=2, seq_len=64, vocab=512, n_batches=4),
        pretraining_loss,
        HParams(lr=1e-3, batch_count=2),
        name="fsdp_gpu",
        save_dir=str(tmp_path),
    )
    params, bt = FSDPExecutor.search(t, [0], 940)
    assert params is not None and bt > 0


def test_bert_and_vit_tiny_train_gpu():
    from saturn_amd.models import get_bert_model, get_vit_model, mlm_loss, vit_loss
    from saturn_amd.models.bert import SyntheticMLM
    from saturn_amd.ops.optim import FusedAdam

    torch.manual_seed(0)
    bert = get_bert_model({"n_layer": 2, "n_embd": 256, "n_head": 4,
                           "vocab_size": 1024, "n_ctx": 128}).to("cuda", torch.bfloat16)
    ds = SyntheticMLM(4, 128, 1024)
    x = torch.stack([ds[i][0] for i in range(4)]).cuda()
    lab = torch.stack([ds[i][1] for i in range(4)]).cuda()
    opt = FusedAdam(bert.parameters(), lr=1e-3)
    loss = mlm_loss(bert(x), lab)
    loss.backward()
    opt.step()
    assert torch.isfinite(torch.tensor(float(loss)))

    vit = get_vit_model({"n_layer": 2, "n_embd": 256, "n_head": 4,
                         "img_size": 64}).to("cuda", torch.bfloat16)
    px = torch.randn(2, 3, 64, 64, device="cuda", dtype=torch.bfloat16)
    y = torch.randint(0, 1000, (2,), device="cuda")
    loss = vit_loss(vit(px), y)
    loss.backward()
    assert torch.isfinite(torch.tensor(float(loss)))


def test_hipgraph_step_matches_eager():
    """A hipGraph-captured step must reproduce the eager param trajectory
    bit-for-bit (same kernels, same order, static buffers)."""
    import copy

    from saturn_amd.models.gptj import get_gptj_model, pretraining_loss
    from saturn_amd.ops.optim import FusedSGD
    from saturn_amd.utils.graph_step import graphed_train_step

    torch.manual_seed(0)
    cfg = {"n_layer": 2, "n_embd": 256, "n_head": 4, "vocab_size": 512,
           "n_ctx": 64, "rotary_dim": 16}
    m1 = get_gptj_model(cfg).to("cuda", torch.bfloat16)
    m2 = copy.deepcopy(m1)
    o1 = FusedSGD(m1.parameters(), lr=1e-3)
    o2 = FusedSGD(m2.parameters(), lr=1e-3)

    g = torch.Generator().manual_seed(9)
    batches = [torch.randint(0, 512, (2, 64), generator=g).cuda()
               for _ in range(3)]

    # eager trajectory (same zero/fwd/bwd/step order as the capture closure)
    eager_losses = []
    for b in batches:
        o1.zero_grad(set_to_none=False)
        loss = pretraining_loss(m1(b), b)
        loss.backward()
        o1.step()
        eager_losses.append(float(loss))

    # warmup mutates m2, so rebuild it identically after capture probing:
    # instead, warm up AND capture on m2, then reset its state to m1's start
    # is invalid — capture bakes pointers.  Correct scheme: warm up on a
    # throwaway copy is impossible (pointers differ), so warm up on m2 and
    # reset the *values* in place before replaying.
    m3 = copy.deepcopy(m2)  # pristine values
    graphed, static_x = graphed_train_step(
        m2, pretraining_loss, o2, batches[0], warmup=2
    )
    with torch.no_grad():
        for p2, p3 in zip(m2.parameters(), m3.parameters()):
            p2.copy_(p3)
    graph_losses = []
    for b in batches:
        static_x.copy_(b)
        graph_losses.append(float(graphed.replay()))

    assert eager_losses == pytest.approx(graph_losses, rel=0, abs=0), (
        eager_losses, graph_losses)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)


def test_native_comm_gang_churn():
    """SURVEY §7 hard-part 3: repeated per-interval communicator
    create/train/destroy cycles must not leak HBM (the reference leaned on
    process death for cleanup; the native engine must clean up
    deliberately)."""
    import torch

    from saturn_amd.comm import create_comm, has_native_comm

    if not has_native_comm():
        import pytest

        pytest.skip("native comm engine not built")
    torch.cuda.synchronize()
    base = None
    for cycle in range(8):
        comm = create_comm(0, 1)
        t = torch.randn(1 << 20, device="cuda")
        comm.all_reduce(t, True)
        comm.broadcast(t, 0)
        comm.join()
        torch.cuda.synchronize()
        del comm, t
        import gc

        gc.collect()
        torch.cuda.synchronize()
        free, _total = torch.cuda.mem_get_info()
        if cycle == 1:
            base = free  # cycle 0 warms allocator/RCCL pools
        elif cycle > 1:
            assert free >= base - (64 << 20), (
                f"HBM leak under comm churn: cycle {cycle} free {free} "
                f"vs base {base}"
            )
