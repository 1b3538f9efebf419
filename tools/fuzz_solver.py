"""Offline solver fuzz: random instances through the full introspection
cycle (solve -> advance bookkeeping -> warm re-solve -> retire tasks ->
re-solve on the restricted incumbent), asserting plan validity each step.

    python tools/fuzz_solver.py [n_instances]

Caught in round 2: Plan.shift clamped start times without consuming the
executed portion of a running task's runtime, so a kept incumbent showed
phantom overlaps (fixed in solver/milp.py; regression test in
tests/test_solver.py::test_shift_shrinks_running_task_runtime).
"""

import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from saturn_amd import HParams, Task
from saturn_amd.core.strategy import Strategy
from saturn_amd.solver.milp import apply_plan, solve


def check_plan_valid(plan, tasks, n_gpus) -> None:
    assert plan.task_names == [t.name for t in tasks]
    for i in range(len(tasks)):
        assert len(plan.gpu_sets[i]) == plan.gpu_counts[i] > 0
        assert all(0 <= g < n_gpus for g in plan.gpu_sets[i])
        assert plan.start_times[i] >= -1e-6
    for i in range(len(tasks)):
        for j in range(i + 1, len(tasks)):
            if set(plan.gpu_sets[i]) & set(plan.gpu_sets[j]):
                si = plan.start_times[i]
                ei = si + plan.runtimes[i]
                sj = plan.start_times[j]
                ej = sj + plan.runtimes[j]
                assert ei <= sj + 1e-4 or ej <= si + 1e-4, (
                    plan.task_names[i], plan.task_names[j], si, ei, sj, ej,
                )


def main() -> None:
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 200
    rng = random.Random(7)
    fails = 0
    for it in range(n):
        n_gpus = rng.choice([1, 2, 4, 8])
        T = rng.randint(1, 6)
        tasks = []
        for k in range(T):
            t = Task(
                lambda: None, lambda: [0] * 10, lambda a, b: None,
                HParams(lr=1e-3, batch_count=rng.randint(5, 200)),
                name=f"t{it}_{k}", save_dir="/tmp/fuzz_solver",
            )
            for g in sorted(
                rng.sample(range(1, n_gpus + 1), rng.randint(1, n_gpus))
            ):
                bt = rng.uniform(0.01, 3.0)
                t.strategies[g] = Strategy(
                    None, g, {"x": 1}, bt * t.total_batches, batch_time=bt
                )
            tasks.append(t)
        try:
            plan = solve(tasks, n_gpus=n_gpus, timeout=6)
            check_plan_valid(plan, tasks, n_gpus)
            apply_plan(tasks, plan)
            for t in tasks:
                t.batches_completed = rng.randint(0, t.total_batches // 2)
            interval = rng.uniform(0.5, 50.0)
            plan2 = solve(
                tasks, presolved=plan, interval=interval, timeout=6,
                n_gpus=n_gpus,
            )
            check_plan_valid(plan2, tasks, n_gpus)
            if T > 1:
                keep = [t for t in tasks if rng.random() > 0.4] or tasks[:1]
                pre = plan2.restrict([t.name for t in keep])
                plan3 = solve(
                    keep, presolved=pre, interval=interval, timeout=6,
                    n_gpus=n_gpus,
                )
                check_plan_valid(plan3, keep, n_gpus)
        except AssertionError as e:
            fails += 1
            print("FAIL", it, repr(e)[:200])
            if fails > 5:
                break
    print(f"done: {n - fails}/{n} clean")
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
