"""LayerNorm kernel microbenchmark at the GPT-J bench shape (8192x4096)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from saturn_amd.ops import require_ext


def main() -> None:
    ext = require_ext()
    rows, cols = (
        int(sys.argv[1]) if len(sys.argv) > 1 else 8192,
        int(sys.argv[2]) if len(sys.argv) > 2 else 4096,
    )
    x = torch.randn(rows, cols, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(cols, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(cols, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn_like(x)
    y, mean, rstd = ext.norm_fwd(x, w, b, 1e-5, False)
    torch.cuda.synchronize()
    for name, fn in [
        ("fwd", lambda: ext.norm_fwd(x, w, b, 1e-5, False)),
        ("bwd", lambda: ext.norm_bwd(dy, x, w, mean, rstd, False, True)),
        ("torch_fwd", lambda: torch.nn.functional.layer_norm(
            x, (cols,), w, b)),
    ]:
        for _ in range(10):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(50):
            fn()
        torch.cuda.synchronize()
        print(f"{name} {(time.perf_counter() - t0) / 50 * 1e6:.1f} us")


if __name__ == "__main__":
    main()
