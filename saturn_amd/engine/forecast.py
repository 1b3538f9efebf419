"""Interval forecasting: which tasks run this interval and for how many
batches.

Parity with reference ``saturn/executor/executor.py:132-178`` (forecast),
with two fixes:

- bookkeeping advances ``task.batches_completed`` instead of destructively
  decrementing every ``strategies[g].runtime`` (reference quirk #5);
- a relevant task always runs at least one batch, so a job whose per-batch
  time exceeds the interval still makes progress instead of looping forever.
"""

from __future__ import annotations

import logging
from typing import List, Set, Tuple

log = logging.getLogger(__name__)


def forecast(task_list: List, interval: float, plan) -> Tuple[List, List[int], Set]:
    """Project one interval forward.

    Returns (relevant_tasks, batches_to_run, completing_tasks).  Also
    advances each relevant task's ``batches_completed`` by its quota so the
    overlapped next-interval solve sees post-interval remaining work
    (the reference mutates strategy runtimes at the same point,
    executor.py:166-172).
    """
    name_to_plan = {nm: i for i, nm in enumerate(plan.task_names)}
    relevant, batches, completing = [], [], set()
    for task in task_list:
        i = name_to_plan[task.name]
        st = plan.start_times[i]
        if st >= interval:
            continue
        strat = task.selected_strategy
        if strat is None:
            raise RuntimeError(f"task {task.name} has no selected strategy")
        if strat.batch_time is not None:
            bt = float(strat.batch_time)
        else:
            bt = float(strat.runtime) / max(1, task.total_batches)
        time_here = interval - st
        quota = int(time_here // bt) if bt > 0 else task.remaining_batches
        quota = max(1, quota)
        quota = min(task.remaining_batches, quota)
        if quota <= 0:
            continue
        relevant.append(task)
        batches.append(quota)
        task.batches_completed += quota
        if task.remaining_batches <= 0:
            completing.add(task)
            log.info("task %s will finish inside this interval", task.name)
    return relevant, batches, completing
