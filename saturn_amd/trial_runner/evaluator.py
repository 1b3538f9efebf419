"""Trial runner: empirically profile every (task, gpu-count, technique) cell.

Parity with reference ``saturn/trial_runner/PerformanceEvaluator.py:21-115``:
fan the cells out over the node's GPUs, call each technique's own
``search()`` (which autotunes its parameters with short timed trials), and
write the winning ``Strategy`` per (task, gpu-count) back onto the tasks.
Differences from the reference:

- cells are packed onto GPUs with a thread-pool + GPU-slot allocator instead
  of Ray placement (``ray_search.options(num_gpus=g)``,
  PerformanceEvaluator.py:74-84);
- each cell runs in its own spawn subprocess with ``HIP_VISIBLE_DEVICES``
  restricted to its gang, so an intentionally-OOMing trial cannot poison a
  sibling (SURVEY §7 hard-part 5);
- search returns per-batch seconds; whole-job runtime and the per-batch time
  are both stored (``Strategy.batch_time``), feeding the solver's
  remaining-work computation.
"""

from __future__ import annotations

import logging
import threading
from timeit import default_timer as timer
from typing import List, Optional

from saturn_amd.core.strategy import INFEASIBLE_RUNTIME, Strategy
from saturn_amd.engine.gang import call_in_subprocess
from saturn_amd.library import retrieve
from saturn_amd.solver.milp import detect_gpu_count

log = logging.getLogger(__name__)


class _GpuPool:
    """Reserve ``g`` concrete GPU ids at a time (any ids — trials are
    placement-agnostic on a fully-connected xGMI node)."""

    def __init__(self, n: int) -> None:
        self.free = set(range(n))
        self.cv = threading.Condition()

    def acquire(self, g: int) -> List[int]:
        with self.cv:
            while len(self.free) < g:
                self.cv.wait()
            ids = sorted(self.free)[:g]
            self.free -= set(ids)
            return ids

    def release(self, ids: List[int]) -> None:
        with self.cv:
            self.free |= set(ids)
            self.cv.notify_all()


def _run_cell(executor, task, g: int, tid: int):
    """Subprocess body for one trial cell."""
    return executor.search(task, list(range(g)), tid)


def search(
    tasks: List,
    executor_names: Optional[List[str]] = None,
    log_level: bool = False,
    n_gpus: Optional[int] = None,
    trial_timeout: Optional[float] = 1800.0,
    isolate: bool = True,
    profile: bool = False,
) -> None:
    """Profile all cells and attach per-gpu-count winning Strategies.

    ``isolate=False`` runs cells in-process (CPU test mode — no HIP context
    to isolate).  ``profile=True`` re-runs each task's fastest cell under
    ``rocprofv3 --kernel-trace --stats`` and attaches the top kernel rows
    plus a rocm-smi sample to that Strategy's parameters (the north star's
    rocprof/rocm-smi-fed trial profiling; the reference used timeit only,
    SURVEY §5.1)."""
    if log_level:
        logging.basicConfig(
            format="%(asctime)s %(levelname)-8s %(message)s",
            level=logging.INFO,
            datefmt="%Y-%m-%d %H:%M:%S",
        )
    executors = retrieve(executor_names)
    if n_gpus is None:
        n_gpus = detect_gpu_count()
    default_range = list(range(1, n_gpus + 1))

    cells = []  # (task_idx, g, exec_idx)
    for ti, t in enumerate(tasks):
        for g in t.gpu_range or default_range:
            if g > n_gpus:
                continue
            for ei in range(len(executors)):
                cells.append((ti, g, ei))

    results = {}
    pool = _GpuPool(n_gpus)
    lock = threading.Lock()

    def run(cell_idx: int) -> None:
        ti, g, ei = cells[cell_idx]
        task, executor = tasks[ti], executors[ei]
        ids = pool.acquire(g)
        t0 = timer()
        try:
            if isolate:
                visible = ",".join(str(i) for i in ids)
                params, bt = call_in_subprocess(
                    _run_cell,
                    executor,
                    task,
                    g,
                    cell_idx,
                    env={
                        "HIP_VISIBLE_DEVICES": visible,
                        "CUDA_VISIBLE_DEVICES": visible,
                    },
                    timeout=trial_timeout,
                )
            else:
                params, bt = executor.search(task, list(range(g)), cell_idx)
        except Exception as e:  # infeasible / crashed cell
            log.info(
                "trial (%s, %dG, %s) failed: %s",
                task.name,
                g,
                getattr(executor, "name", executor),
                e,
            )
            params, bt = None, float("inf")
        finally:
            pool.release(ids)
        with lock:
            results[(ti, g, ei)] = (params, bt)
        log.info(
            "trial (%s, %dG, %s): params=%s batch_time=%.4fs (%.1fs trial)",
            task.name,
            g,
            getattr(executor, "name", executor),
            params,
            bt if bt == bt else -1.0,
            timer() - t0,
        )

    log.info("%d trial cells to run on %d GPUs", len(cells), n_gpus)
    threads = [
        threading.Thread(target=run, args=(i,), daemon=True)
        for i in range(len(cells))
    ]
    for th in threads:
        th.start()
    for th in threads:
        th.join()

    # Initialize every cell with the infeasible sentinel, then pick winners
    # (reference PerformanceEvaluator.py:96-115).
    for t in tasks:
        for g in default_range:
            t.strategies[g] = Strategy(None, g, None, INFEASIBLE_RUNTIME)
    for ti, t in enumerate(tasks):
        for g in t.gpu_range or default_range:
            if g > n_gpus:
                continue
            best = None  # (bt, executor, params)
            for ei, ex in enumerate(executors):
                params, bt = results.get((ti, g, ei), (None, float("inf")))
                if params is not None and bt == bt and bt != float("inf"):
                    if best is None or bt < best[0]:
                        best = (bt, ex, params)
            if best is not None:
                bt, ex, params = best
                t.strategies[g] = Strategy(
                    ex,
                    g,
                    params,
                    runtime=bt * t.total_batches,
                    batch_time=bt,
                )

    if profile:
        _profile_winners(tasks, n_gpus)


def _profile_winners(tasks: List, n_gpus: int) -> None:
    """Re-run each task's fastest feasible cell under rocprofv3 and attach
    kernel-time + rocm-smi evidence to its Strategy parameters."""
    import os
    import tempfile

    import dill

    from saturn_amd.trial_runner.profiler import (
        rocm_smi_sample,
        rocprof_stats,
    )

    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False

    for ti, t in enumerate(tasks):
        feas = [
            (s.batch_time, g, s)
            for g, s in t.strategies.items()
            if s is not None and s.feasible and s.batch_time is not None
        ]
        if not feas:
            continue
        _, g, strat = min(feas)
        if strat.parameters is None:
            continue
        if not has_gpu:
            continue  # rocprofv3 needs a device; CPU runs keep timeit-only
        with tempfile.TemporaryDirectory() as td:
            payload = os.path.join(td, "cell.pkl")
            result = os.path.join(td, "out.pkl")
            with open(payload, "wb") as fh:
                dill.dump((strat.executor, t, g, 10_000 + ti), fh)
            import sys as _sys

            # Bounded: rocprofv3 tracing a cell that itself gang-spawns
            # workers can fail to drain (observed on hardware) — a timeout
            # degrades this task to timeit-only instead of wedging search
            rows = rocprof_stats(
                [_sys.executable, "-m", "saturn_amd.trial_runner.cell_main",
                 payload, result],
                timeout=240.0,
            )
        if rows:
            strat.parameters["kernels"] = [
                {
                    "name": (r.get("Name") or r.get("KernelName") or "?")[:100],
                    "total_ns": r.get("TotalDurationNs")
                    or r.get("DurationNs"),
                    "calls": r.get("Calls") or r.get("TotalCalls"),
                }
                for r in rows[:10]
            ]
        smi = rocm_smi_sample()
        if smi:
            strat.parameters["rocm_smi"] = smi[: max(1, g)]
        log.info(
            "profiled winner for %s at %dG: %d kernel rows",
            t.name, g, len(rows or []),
        )
