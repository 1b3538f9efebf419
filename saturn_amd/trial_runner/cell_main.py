"""CLI body for one profiled trial cell.

``search()`` re-runs a task's winning cell under ``rocprofv3`` to attach a
measured kernel table to its Strategy (the north star's rocprof-fed
profiling; the reference only ever used timeit —
PerformanceEvaluator.py:21-30).  rocprofv3 wraps a whole process, so the
cell needs a process entry point:

    python -m saturn_amd.trial_runner.cell_main payload.pkl result.pkl

payload.pkl (dill): (executor_cls, task, n_gpus, tid)
result.pkl  (dill): (params, batch_time)
"""

from __future__ import annotations

import sys

import dill


def main() -> None:
    payload_path, result_path = sys.argv[1], sys.argv[2]
    with open(payload_path, "rb") as fh:
        executor, task, g, tid = dill.load(fh)
    out = executor.search(task, list(range(g)), tid)
    with open(result_path, "wb") as fh:
        dill.dump(out, fh)


if __name__ == "__main__":
    main()
