"""Attention kernel microbenchmark: TF/s of attn_fwd / attn_bwd vs shapes.

    python tools/attn_bench.py            # default shape sweep
Prints one line per shape: fwd/bwd time and effective TFLOP/s (causal
halves the FLOPs).  Within-run interleaved repeats (guide §5.4 rule 24).
"""

import sys
import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from timeit import default_timer as timer

import torch

from saturn_amd.ops import require_ext


def bench_shape(B, H, T, D, iters=20):
    ext = require_ext()
    torch.manual_seed(0)
    q = torch.randn(B, H, T, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    o, lse = ext.attn_fwd(q, k, v, True)
    do = torch.randn_like(o)
    torch.cuda.synchronize()

    t0 = timer()
    for _ in range(iters):
        o, lse = ext.attn_fwd(q, k, v, True)
    torch.cuda.synchronize()
    t_fwd = (timer() - t0) / iters

    t0 = timer()
    for _ in range(iters):
        dq, dk, dv = ext.attn_bwd(do, q, k, v, o, lse, True)
    torch.cuda.synchronize()
    t_bwd = (timer() - t0) / iters

    flops_fwd = 2 * 2 * B * H * T * T * D * 0.5  # causal
    flops_bwd = flops_fwd * 2.5
    print(
        f"B{B} H{H} T{T} D{D}: fwd {t_fwd*1e6:8.1f}us {flops_fwd/t_fwd/1e12:7.1f} TF | "
        f"bwd {t_bwd*1e6:8.1f}us {flops_bwd/t_bwd/1e12:7.1f} TF",
        flush=True,
    )


def burn_in():
    """Ramp clocks before measuring (DVFS: first-measured shape reads low)."""
    ext = require_ext()
    q = torch.randn(8, 16, 1024, 128, device="cuda", dtype=torch.bfloat16)
    for _ in range(50):
        ext.attn_fwd(q, q, q, True)
    torch.cuda.synchronize()


if __name__ == "__main__":
    burn_in()
    for shape in [
        (8, 16, 512, 256),    # GPT-J bench shape
        (16, 16, 512, 256),
        (4, 32, 2048, 128),   # llama-8B-ish
        (16, 64, 2048, 128),  # guide's attention ladder shape
        (8, 12, 1024, 64),    # gpt2
    ]:
        bench_shape(*shape)
