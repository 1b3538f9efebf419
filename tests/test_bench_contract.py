"""Guard the driver's bench.py contract: the round-end scale run launches
`python -m torch.distributed.run --nproc-per-node N bench.py --gpus N ...`
one rank per GPU.  This must work first try on the 8-GPU node, so the
rendezvous + sharding + MAX-over-ranks + single-JSON-line contract is
exercised here on CPU/gloo at world 2."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_torchrun_world2_gloo():
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29539",
            "bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
            "--layers", "2", "--seq", "32", "--batch", "2",
        ],
        cwd=REPO,
        env=env,
        capture_output=True,
        text=True,
        timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    # exactly one JSON line on stdout, from rank 0
    json_lines = [
        l for l in out.stdout.splitlines() if l.startswith("{")
    ]
    assert len(json_lines) == 1, out.stdout
    d = json.loads(json_lines[0])
    assert d["n_gpus"] == 2
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["metric"] == "samples_per_s" and d["value"] > 0
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 4  # per-GPU batch x world


def test_bench_single_process_defaults():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--layers", "1", "--seq", "32", "--batch", "2"],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    d = json.loads(
        [l for l in out.stdout.splitlines() if l.startswith("{")][0]
    )
    assert d["n_gpus"] == 1 and d["higher_is_better"] is True
