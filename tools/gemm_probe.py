"""Probe the GPT-J step's GEMM shapes: stock vs TunableOp-tuned algos.

Usage (on the GPU box):
    python tools/gemm_probe.py base          # stock dispatch
    python tools/gemm_probe.py read          # read tuned csv, tuning off
    python tools/gemm_probe.py tune out.csv  # re-tune with a long budget

Prints per-shape ms and effective TFLOP/s for forward (TN), dgrad (NN)
and wgrad (NT) of every linear in the GPT-J-6B step at M = B*T = 8192.
"""

from __future__ import annotations

import os
import sys

M = 8192
SHAPES = [
    ("proj_4096", 4096, 4096),
    ("fc_in", 16384, 4096),
    ("fc_out", 4096, 16384),
    ("lm_head", 50400, 4096),
]


def main() -> None:
    mode = sys.argv[1] if len(sys.argv) > 1 else "base"
    csv = sys.argv[2] if len(sys.argv) > 2 else "tools/tunableop_gfx950.csv"

    if mode == "tune":
        os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
        os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
        os.environ["PYTORCH_TUNABLEOP_FILENAME"] = csv
        os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS", "200")
        os.environ.setdefault("PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS", "300")

    import torch
    import torch.nn.functional as F

    if mode == "read":
        import torch.cuda.tunable as tunable

        tunable.enable(True)
        tunable.tuning_enable(False)
        tunable.read_file(csv)
        print(f"loaded {len(tunable.get_results())} tuned entries")

    torch.manual_seed(0)
    dev = "cuda"

    def bench(fn, iters=30):
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        for _ in range(iters):
            fn()
        e.record()
        torch.cuda.synchronize()
        return s.elapsed_time(e) / iters

    print(f"mode={mode}  M={M}")
    tot = {"fwd": 0.0, "dgrad": 0.0, "wgrad": 0.0}
    for name, n, k in SHAPES:
        x = torch.randn(M, k, device=dev, dtype=torch.bfloat16)
        w = torch.randn(n, k, device=dev, dtype=torch.bfloat16)
        b = torch.randn(n, device=dev, dtype=torch.bfloat16)
        dy = torch.randn(M, n, device=dev, dtype=torch.bfloat16)
        fl = 2.0 * M * n * k / 1e12

        t_f = bench(lambda: F.linear(x, w, b))
        t_d = bench(lambda: dy @ w)
        t_w = bench(lambda: dy.t() @ x)
        tot["fwd"] += t_f
        tot["dgrad"] += t_d
        tot["wgrad"] += t_w
        print(
            f"{name:10s} fwd {t_f:7.3f} ms {fl/t_f*1e3:7.0f} TF | "
            f"dgrad {t_d:7.3f} ms {fl/t_d*1e3:7.0f} TF | "
            f"wgrad {t_w:7.3f} ms {fl/t_w*1e3:7.0f} TF"
        )
    print(
        f"per-layer-set totals: fwd {tot['fwd']:.3f} dgrad {tot['dgrad']:.3f} "
        f"wgrad {tot['wgrad']:.3f} ms"
    )

    if mode == "tune":
        import torch.cuda.tunable as tunable

        tunable.write_file(csv)
        print(f"wrote {csv}")


if __name__ == "__main__":
    main()
