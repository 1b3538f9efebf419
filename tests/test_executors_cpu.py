"""Pipeline and Spilled executors on CPU, plus the pipeline partitioner."""

import pytest
import torch
import torch.nn as nn

from saturn_amd import HParams, Strategy, Task
from saturn_amd.executors.pipeline import PipelineExecutor
from saturn_amd.executors.spilled import SpilledExecutor
from saturn_amd.models import get_mlp_dataloader, get_mlp_model, mse_loss
from saturn_amd.models.gptj import (
    as_sequential,
    get_gptj_model,
    make_token_dataloader,
    pretraining_loss,
)
from saturn_amd.parallel.pipeline import PipelinedModel, balance_by_params


def test_balance_by_params():
    seq = nn.Sequential(*[nn.Linear(16, 16) for _ in range(10)])
    for n in (2, 3, 4):
        bal = balance_by_params(seq, n)
        assert sum(bal) == 10 and len(bal) == n
        assert all(b >= 1 for b in bal)


def test_pipelined_model_matches_unpipelined():
    torch.manual_seed(0)
    mk = lambda: get_gptj_model(
        {"n_layer": 4, "n_embd": 64, "n_head": 2, "vocab_size": 128,
         "n_ctx": 32, "rotary_dim": 8}
    )
    ref = mk()
    m = mk()
    seq = as_sequential(m)
    pipe = PipelinedModel(seq, ["cpu", "cpu"], chunks=2)
    x = torch.randint(0, 128, (4, 32))
    l_ref = pretraining_loss(ref(x), x)
    l_pipe = pretraining_loss(pipe(x), x)
    assert abs(l_ref.item() - l_pipe.item()) < 1e-4
    l_pipe.backward()
    g = next(p.grad for p in pipe.parameters() if p.grad is not None)
    assert torch.isfinite(g).all()


def gptj_task(name, save_dir, batch_count=4):
    return Task(
        lambda kwargs=None: get_gptj_model(
            {"n_layer": 4, "n_embd": 64, "n_head": 2, "vocab_size": 128,
             "n_ctx": 32, "rotary_dim": 8}
        ),
        make_token_dataloader(batch_size=2, seq_len=32, vocab=128, n_batches=8),
        pretraining_loss,
        HParams(lr=1e-3, batch_count=batch_count),
        name=name,
        save_dir=save_dir,
    )


def test_pipeline_executor_search_and_execute(save_dir):
    t = gptj_task("pipe_t", save_dir)
    params, bt = PipelineExecutor.search(t, [0, 1], 920)
    assert params is not None and "chunks" in params and bt > 0
    t.strategies[2] = Strategy(PipelineExecutor, 2, params, bt * 4, batch_time=bt)
    t.select_strategy(t.strategies[2])
    PipelineExecutor.execute(t, [0, 1], 920, 2)
    assert t.has_ckpt()


def test_pipeline_rejects_single_gpu(save_dir):
    t = gptj_task("pipe_1g", save_dir)
    params, bt = PipelineExecutor.search(t, [0], 921)
    assert params is None


def test_spilled_executor_search_and_execute(save_dir):
    t = gptj_task("spill_t", save_dir)
    params, bt = SpilledExecutor.search(t, [0], 922)
    assert params is not None and "partitions" in params
    t.strategies[1] = Strategy(SpilledExecutor, 1, params, bt * 4, batch_time=bt)
    t.select_strategy(t.strategies[1])
    SpilledExecutor.execute(t, [0], 922, 2)
    assert t.has_ckpt()


def test_spilled_rejects_multi_gpu(save_dir):
    t = gptj_task("spill_2g", save_dir)
    params, bt = SpilledExecutor.search(t, [0, 1], 923)
    assert params is None


def test_llama_tiny_cpu_trains():
    from saturn_amd.models.llama import get_llama_model, llama_loss

    torch.manual_seed(0)
    m = get_llama_model({"preset": "8b", "n_layer": 2, "n_ctx": 32,
                         "vocab_size": 256})
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    x = torch.randint(0, 256, (2, 32))
    l0 = None
    for i in range(5):
        loss = llama_loss(m(x), x)
        loss.backward()
        opt.step()
        opt.zero_grad()
        if i == 0:
            l0 = loss.item()
    assert loss.item() < l0


# ---------------------------------------------------------------------------
# Property-based: bucket/shard partition invariants on random model shapes
# ---------------------------------------------------------------------------
from hypothesis import given, settings, strategies as st  # noqa: E402


@given(
    st.lists(st.integers(1, 300), min_size=1, max_size=12),
    st.floats(0.001, 1.0),
)
@settings(max_examples=20, deadline=None)
def test_ddp_buckets_partition_params_exactly(sizes, bucket_mb):
    import torch

    from saturn_amd.parallel.ddp import BucketedDDP

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            for i, n in enumerate(sizes):
                setattr(self, f"p{i}", torch.nn.Parameter(torch.randn(n)))

    m = M()
    # world 1 skips buckets by default (no comm to feed); force them here
    # to exercise the partitioning logic the world>1 path uses
    ddp = BucketedDDP(m, bucket_mb=bucket_mb, use_buckets=True)
    seen = set()
    for b in ddp.buckets:
        for p in b.params:
            assert id(p) not in seen, "param in two buckets"
            seen.add(id(p))
    assert seen == {id(p) for p in m.parameters()}
    # grads are views into the flat buffer: writing the buffer is visible
    for b in ddp.buckets:
        b.flat.fill_(3.0)
    for p in m.parameters():
        assert p.grad is not None and torch.all(p.grad == 3.0)
    ddp.zero_grad_buffers()
    for p in m.parameters():
        assert torch.all(p.grad == 0.0)


@given(st.integers(1, 6), st.integers(1, 4))
@settings(max_examples=10, deadline=None)
def test_zero3_units_cover_model(n_layer, heads):
    import torch

    from saturn_amd.models.gptj import get_gptj_model
    from saturn_amd.parallel.zero3 import Zero3Model

    torch.manual_seed(0)
    m = get_gptj_model({"n_layer": n_layer, "n_embd": 16 * heads,
                        "n_head": heads, "vocab_size": 64, "n_ctx": 16,
                        "rotary_dim": 8})
    total = sum(p.numel() for p in m.parameters())
    z3 = Zero3Model(m, prefetch=False)
    unit_total = sum(int(u.numels.sum()) if hasattr(u.numels, "sum")
                     else sum(u.numels) for u in z3.units)
    assert unit_total == total, (unit_total, total)
    x = torch.randint(0, 64, (1, 16))
    from saturn_amd.models.gptj import pretraining_loss

    loss = pretraining_loss(z3(x), x)
    loss.backward()
    z3.grad_sync()
    assert all(u.shard.grad is not None for u in z3.units)


@given(
    st.lists(st.integers(1, 200), min_size=2, max_size=16),
    st.integers(2, 4),
)
@settings(max_examples=20, deadline=None)
def test_pipeline_balance_covers_all_layers(sizes, n_stages):
    import torch

    from saturn_amd.parallel.pipeline import balance_by_params

    if len(sizes) < n_stages:
        sizes = sizes * n_stages
    seq = torch.nn.Sequential(*[torch.nn.Linear(n, n) for n in sizes])
    bal = balance_by_params(seq, n_stages)
    assert len(bal) == n_stages
    assert all(b >= 1 for b in bal), bal
    assert sum(bal) == len(seq), (bal, len(seq))


def test_balance_by_time_vs_params():
    """Time balance must react to uneven layer cost where parameter
    balance cannot (reference balance_by_time, Pipeline.py:94-103)."""
    import torch
    import torch.nn as nn

    from saturn_amd.parallel.pipeline import (
        balance_by_params,
        balance_by_time,
    )

    class _Rep(nn.Module):
        def __init__(self, reps):
            super().__init__()
            self.reps = reps

        def forward(self, x):
            for _ in range(self.reps):
                x = torch.tanh(x)
            return x

    seq = nn.Sequential(_Rep(400), _Rep(1), _Rep(1), _Rep(1))
    assert balance_by_params(seq, 2) == [2, 2]  # no params -> naive split
    bal = balance_by_time(seq, torch.randn(256, 256), 2)
    assert sum(bal) == 4 and len(bal) == 2
    assert bal[0] == 1, bal  # the slow layer dominates stage 0
