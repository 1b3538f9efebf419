"""Flagship benchmark: GPT-J-6B bf16 data-parallel training step.

This measures the reference's headline workload (GPT-J-6B
fine-tune, ctx 512, batch 8 per GPU, SGD — simple-verification.py:58-73) on
saturn_amd's own training stack: fused CDNA4 kernels for
LayerNorm/RoPE/attention/CE/optimizer, bucketed flat-buffer DDP over RCCL.

Contract (driver):
    python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches via torch.distributed.run; ranks read
RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* from the env.  Rank 0 prints ONE JSON
line; `value` is whole-job samples/sec aggregated over all N GPUs;
per-GPU work is fixed (weak scaling).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
from timeit import default_timer as timer

import torch


def log(*a):
    print(*a, file=sys.stderr, flush=True)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch", type=int, default=16, help="per-GPU batch size")
    ap.add_argument("--seq", type=int, default=512)
    ap.add_argument("--layers", type=int, default=None,
                    help="override layer count (default: model standard)")
    ap.add_argument("--model", type=str, default="gptj-6b")
    ap.add_argument("--bucket-mb", type=float, default=64.0)
    ap.add_argument(
        "--hipgraph",
        action="store_true",
        help="capture the step in a hipGraph and replay it (world 1 only)",
    )
    ap.add_argument(
        "--tunableop",
        action="store_true",
        help="use the offline-tuned hipBLASLt/rocBLAS GEMM algo table "
        "(tools/tunableop_gfx950.csv), tuning disabled at runtime",
    )
    args = ap.parse_args()

    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        raise SystemExit(
            "--gpus N>1 must be launched via torch.distributed.run "
            "(one rank per GPU); a single process would hang at rendezvous."
        )
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_gpu = torch.cuda.is_available()
    device = torch.device("cuda", local_rank) if use_gpu else torch.device("cpu")
    dtype = torch.bfloat16 if use_gpu else torch.float32

    import torch.distributed as dist

    if world > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29517")
        dist.init_process_group(
            backend="nccl" if use_gpu else "gloo", rank=rank, world_size=world
        )
    if use_gpu:
        torch.cuda.set_device(local_rank)
        # the bench must run on our kernels, never a silent fallback
        from saturn_amd.ops import require_ext

        require_ext()
        if args.tunableop:
            # READ-ONLY use of the offline-tuned GEMM algo table; runtime
            # tuning stays off so the timed region is never polluted
            csv = os.path.join(
                os.path.dirname(os.path.abspath(__file__)),
                "tools", "tunableop_gfx950.csv",
            )
            if os.path.isfile(csv):
                import torch.cuda.tunable as tunable

                tunable.enable(True)
                tunable.tuning_enable(False)
                tunable.read_file(csv)
                log(f"[rank {rank}] tunableop: {len(tunable.get_results())} tuned GEMMs loaded")
            else:
                log(f"[rank {rank}] tunableop csv missing at {csv}; skipping")

    from saturn_amd.ops.optim import FusedSGD
    from saturn_amd.parallel.ddp import BucketedDDP

    torch.manual_seed(1234)
    log(f"[rank {rank}] building {args.model} on {device}...")
    # build directly on the device IN the compute dtype: at world 8 a
    # host-side build would transiently hold 8 x 24 GB fp32 replicas in
    # DRAM, and fp32-on-device doubles the init writes
    prev_dtype = torch.get_default_dtype()
    if use_gpu:
        torch.set_default_dtype(dtype)
    with device:
        if args.model == "llama-3-8b":
            from saturn_amd.models.llama import (
                LlamaConfig,
                LlamaForCausalLM,
                llama_loss as loss_fn,
            )

            mcfg = LlamaConfig(n_ctx=max(args.seq, 2048))
            if args.layers is not None:
                from dataclasses import replace

                mcfg = replace(mcfg, n_layer=args.layers)
            model = LlamaForCausalLM(mcfg)
            vocab = mcfg.vocab_size
        else:
            from saturn_amd.models.gptj import (
                GPTJConfig,
                GPTJForCausalLM,
                pretraining_loss as loss_fn,
            )

            mcfg = GPTJConfig(
                n_layer=args.layers if args.layers is not None else 28,
                n_ctx=args.seq,
            )
            model = GPTJForCausalLM(mcfg)
            vocab = mcfg.vocab_size
    torch.set_default_dtype(prev_dtype)
    model = model.to(dtype=dtype)
    model.train()
    n_params = sum(p.numel() for p in model.parameters())
    ddp = BucketedDDP(model, bucket_mb=args.bucket_mb)
    opt = FusedSGD(model.parameters(), lr=1e-5)

    g = torch.Generator(device="cpu").manual_seed(4321 + rank)
    x = torch.randint(0, vocab, (args.batch, args.seq), generator=g).to(device)

    if args.hipgraph:
        if world > 1:
            raise SystemExit("--hipgraph is single-GPU only this round")
        if not use_gpu:
            raise SystemExit("--hipgraph needs a GPU (hipGraph capture)")
        from saturn_amd.utils.graph_step import graphed_train_step

        graphed, static_x = graphed_train_step(ddp.module, loss_fn, opt, x,
                                               ddp=ddp)
        static_x.copy_(x)

        def step() -> None:
            graphed.replay()

    else:

        def step() -> None:
            logits = ddp(x)
            loss = loss_fn(logits, x)
            loss.backward()
            ddp.grad_sync()
            opt.step()
            ddp.zero_grad_buffers()

    log(f"[rank {rank}] {n_params/1e9:.2f}B params; warmup {args.warmup} steps")
    for _ in range(args.warmup):
        step()

    if world > 1:
        dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = timer()
    for _ in range(args.steps):
        step()
    if use_gpu:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    t1 = timer()

    elapsed = t1 - t0
    # MAX over ranks
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device if use_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        ms_per_step = elapsed / args.steps * 1000.0
        samples_per_s = args.batch * world * args.steps / elapsed
        print(
            json.dumps(
                {
                    "metric": "samples_per_s",
                    "value": samples_per_s,
                    "unit": "samples/s",
                    "n_gpus": world,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": ms_per_step,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "bf16" if use_gpu else "fp32",
                    "data": "synthetic",
                    "config": {
                        "model": args.model,
                        "n_params_b": round(n_params / 1e9, 2),
                        "global_batch": args.batch * world,
                        "seq_len": args.seq,
                        "parallelism": f"dp{world}",
                        "optimizer": "fused_sgd",
                    },
                }
            ),
            flush=True,
        )
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
