"""The interval-based introspective orchestrator.

Parity with reference ``saturn/orchestrator.py:32-75``: solve up front, then
loop {forecast -> launch next solve overlapped -> execute interval -> adopt
plan}; re-solving between intervals lets the plan adapt as tasks finish.
Fixes the reference's first-call argument-order bug (orchestrator.py:55
passes ``gurobi=interval``) by using keyword-only solver
arguments, and overlaps the solve with a thread instead of a Ray task.
"""

from __future__ import annotations

import logging
from concurrent.futures import ThreadPoolExecutor
from typing import List, Optional

from saturn_amd.engine import execute, forecast
from saturn_amd.solver import Plan, apply_plan, detect_gpu_count, solve

log = logging.getLogger(__name__)


def orchestrate(
    task_list: List,
    log_level: bool = False,
    interval: float = 1000.0,
    n_gpus: Optional[int] = None,
    solver_timeout: Optional[float] = None,
    hysteresis: float = 500.0,
    launch_timeout: Optional[float] = None,
) -> None:
    """Run a batch of profiled tasks to completion.

    Parameters mirror the reference ``orchestrate(task_list, log, interval,
    gurobi)``; ``gurobi`` is gone (HiGHS is built in) and ``n_gpus`` pins the
    node size for CPU-only tests.
    """
    if log_level:
        logging.basicConfig(
            format="%(asctime)s %(levelname)-8s %(message)s",
            level=logging.INFO,
            datefmt="%Y-%m-%d %H:%M:%S",
        )
    if n_gpus is None:
        n_gpus = detect_gpu_count()
    if solver_timeout is None:
        solver_timeout = max(1.0, interval / 2)

    task_list = list(task_list)
    plan: Plan = solve(
        task_list,
        None,
        interval=interval,
        timeout=solver_timeout,
        n_gpus=n_gpus,
        hysteresis=hysteresis,
    )
    apply_plan(task_list, plan)

    pool = ThreadPoolExecutor(max_workers=1)
    try:
        while task_list:
            relevant, batches, completing = forecast(task_list, interval, plan)
            log.info(
                "interval: running %s; expecting %s to finish",
                [t.name for t in relevant],
                [t.name for t in completing],
            )
            next_tasks = [t for t in task_list if t not in completing]
            presolved = plan.restrict([t.name for t in next_tasks]) if next_tasks else None
            fut = (
                pool.submit(
                    solve,
                    next_tasks,
                    presolved,
                    interval=interval,
                    timeout=solver_timeout,
                    n_gpus=n_gpus,
                    hysteresis=hysteresis,
                )
                if next_tasks
                else None
            )
            execute(
                relevant,
                batches,
                interval,
                plan,
                launch_timeout=launch_timeout,
            )
            task_list = next_tasks
            if fut is not None:
                plan = fut.result()
                apply_plan(task_list, plan)
    finally:
        pool.shutdown(wait=False)
