"""Round-2 regression tests for the round-1 advisor findings.

- FusedSGD momentum buffers must be fp32 regardless of param dtype
  (the HIP kernel contract; advisor finding #1);
- fp32 master-weight mode must rescue sub-resolution bf16 updates;
- BucketedDDP must reduce a bucket whose params received no grad
  (advisor finding #2: un-reduced buckets silently diverged before);
- Zero3 must actually free gathered parameter storage after a unit's
  forward (advisor finding #3: saved autograd views kept every unit's
  full buffer alive).
"""

import torch
import torch.nn as nn

from saturn_amd.executors.launch import gang_spawn
from saturn_amd.ops.optim import FusedAdam, FusedSGD


# ---------------------------------------------------------------------------
# FusedSGD / FusedAdam precision contract
# ---------------------------------------------------------------------------
def test_fused_sgd_momentum_buffers_are_fp32():
    p = nn.Parameter(torch.randn(16, dtype=torch.bfloat16))
    opt = FusedSGD([p], lr=0.1, momentum=0.9)
    p.grad = torch.randn_like(p)
    opt.step()
    buf = opt.state[p]["momentum_buffer"]
    assert buf.dtype == torch.float32


def test_fused_sgd_momentum_matches_fp32_reference():
    torch.manual_seed(0)
    init = torch.randn(64)
    g = [torch.randn(64) for _ in range(5)]

    p32 = nn.Parameter(init.clone())
    ref = torch.optim.SGD([p32], lr=0.05, momentum=0.9, weight_decay=0.01)
    pf = nn.Parameter(init.clone())
    ours = FusedSGD([pf], lr=0.05, momentum=0.9, weight_decay=0.01)
    for gi in g:
        p32.grad = gi.clone()
        ref.step()
        pf.grad = gi.clone()
        ours.step()
    # torch SGD applies wd into the grad before momentum, same as ours
    assert torch.allclose(p32.data, pf.data, atol=1e-5), (
        (p32.data - pf.data).abs().max()
    )


def test_fused_sgd_master_weights_rescue_small_updates():
    # lr*grad = 1e-4 is far below bf16 resolution at 1.0 (~3.9e-3): without
    # a master copy every update cancels; with one they accumulate.
    steps, lr = 20, 1e-4
    p_plain = nn.Parameter(torch.ones(8, dtype=torch.bfloat16))
    plain = FusedSGD([p_plain], lr=lr)
    p_master = nn.Parameter(torch.ones(8, dtype=torch.bfloat16))
    master = FusedSGD([p_master], lr=lr, master_weights=True)
    for _ in range(steps):
        p_plain.grad = torch.ones_like(p_plain)
        plain.step()
        p_master.grad = torch.ones_like(p_master)
        master.step()
    assert torch.equal(p_plain.data, torch.ones_like(p_plain))  # cancelled
    mw = master.state[p_master]["master"]
    assert mw.dtype == torch.float32
    assert torch.allclose(mw, torch.full((8,), 1.0 - steps * lr), atol=1e-6)


def test_fused_adam_master_weights_track_fp32():
    torch.manual_seed(1)
    init = torch.randn(32)
    g = [torch.randn(32) for _ in range(4)]

    p32 = nn.Parameter(init.clone())
    ref = FusedAdam([p32], lr=1e-2, weight_decay=0.01)
    pb = nn.Parameter(init.clone().to(torch.bfloat16))
    ours = FusedAdam([pb], lr=1e-2, weight_decay=0.01, master_weights=True)
    for gi in g:
        p32.grad = gi.clone()
        ref.step()
        pb.grad = gi.clone().to(torch.bfloat16)
        ours.step()
    mw = ours.state[pb]["master"]
    # master differs from the fp32 run only by bf16 gradient rounding
    assert torch.allclose(mw, p32.data, atol=3e-2), (mw - p32.data).abs().max()
    assert torch.equal(pb.data, mw.to(torch.bfloat16))


# ---------------------------------------------------------------------------
# BucketedDDP with a grad-less parameter in a bucket
# ---------------------------------------------------------------------------
class _PartialModel(nn.Module):
    def __init__(self):
        super().__init__()
        self.used = nn.Linear(4, 4, bias=False)
        self.unused = nn.Linear(4, 4, bias=False)  # no grad this step

    def forward(self, x):
        return self.used(x)


def _ddp_unused_worker(rank: int, world: int, _):
    import torch

    from saturn_amd.executors.launch import (
        destroy_process_group,
        init_process_group,
    )
    from saturn_amd.parallel.ddp import BucketedDDP

    init_process_group(rank, world)
    try:
        torch.manual_seed(0)
        m = _PartialModel()
        ddp = BucketedDDP(m, bucket_mb=64.0)  # both linears share one bucket
        x = torch.full((2, 4), float(rank + 1))
        loss = ddp(x).sum()
        loss.backward()
        ddp.grad_sync()  # must launch the partially-filled bucket
        if rank == 0:
            return (
                m.used.weight.grad.clone(),
                m.unused.weight.grad.clone(),
                m.used.weight.data.clone(),
            )
        return None
    finally:
        destroy_process_group()


def test_ddp_bucket_with_unused_param_still_reduces():
    got_used, got_unused, w0 = gang_spawn(
        _ddp_unused_worker, 2, 931, None, timeout=300
    )
    # expected grad = mean over ranks of d(sum(W x_r))/dW = mean_r 1^T x_r
    torch.manual_seed(0)
    ref = _PartialModel()
    ref.used.weight.data.copy_(w0)
    acc = torch.zeros_like(ref.used.weight)
    for r in range(2):
        ref.zero_grad()
        x = torch.full((2, 4), float(r + 1))
        ref(x).sum().backward()
        acc += ref.used.weight.grad
    acc /= 2
    assert torch.allclose(got_used, acc, atol=1e-6), (got_used, acc)
    assert torch.equal(got_unused, torch.zeros_like(got_unused))


# ---------------------------------------------------------------------------
# Zero3 storage is actually freed between forward and backward
# ---------------------------------------------------------------------------
def test_zero3_frees_storage_after_forward():
    from saturn_amd.models.gptj import get_gptj_model, pretraining_loss
    from saturn_amd.parallel.zero3 import Zero3Model

    torch.manual_seed(0)
    m = get_gptj_model(
        {"n_layer": 2, "n_embd": 64, "n_head": 2, "vocab_size": 128,
         "n_ctx": 32, "rotary_dim": 8}
    )
    z3 = Zero3Model(m, prefetch=False)
    x = torch.randint(0, 128, (2, 32))
    out = z3(x)
    # every block unit must have RESIZED its gather buffer to 0 bytes even
    # though autograd saved weight views into it (they share the storage)
    for u in z3.units[:-1]:
        assert u.full is not None
        assert u.full.untyped_storage().size() == 0, u.idx
    # backward re-gathers into the same storage; training still exact
    loss = pretraining_loss(out, x)
    loss.backward()
    z3.grad_sync()
    for u in z3.units:
        assert u.shard.grad is not None
        assert torch.isfinite(u.shard.grad).all()
    # after backward everything is freed again
    for u in z3.units:
        assert u.full is None or u.full.untyped_storage().size() == 0
