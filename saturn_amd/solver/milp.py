"""Joint strategy-selection + GPU-apportionment + gang-schedule MILP.

Capability parity with the reference solver (``saturn/solver/milp.py:23-513``)
— same decision structure (per-task strategy selection, GPU occupancy,
start times, pairwise ordering, makespan objective, plan-swap hysteresis) —
reformulated for HiGHS via ``scipy.optimize.milp`` (no Gurobi/PuLP/Ray
dependency) and for a single 8-GPU MI355X node:

- one continuous start time per task instead of the reference's per-(node,
  GPU, task) integer start grid with same-start coupling constraints
  (milp.py:139-149, 233-256) — smaller model, same semantics;
- big-M is a computed schedule horizon instead of the fixed 1e10
  (milp.py:163), which HiGHS handles far better numerically;
- the strategy-column <-> task.strategies mapping is explicit in the Plan
  instead of relying on dict insertion order (reference quirk,
  milp.py:72-81, 478-486);
- a greedy earliest-finish fallback guarantees a valid plan even if the MILP
  times out with no incumbent.

The introspection contract is kept: ``solve(task_list, presolved=...)``
adopts a new plan only when it beats the saved one by more than
``interval + hysteresis`` seconds (reference milp.py:363-381), otherwise the
saved plan is shifted forward by one interval (milp.py:434-442).
"""

from __future__ import annotations

import logging
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
from scipy.optimize import Bounds, LinearConstraint, milp
from scipy.sparse import csr_matrix

from saturn_amd.core.strategy import INFEASIBLE_RUNTIME

log = logging.getLogger(__name__)


# ---------------------------------------------------------------------------
# Plan: the solver's output (the reference's in-memory 6-tuple, milp.py:445,
# made explicit).
# ---------------------------------------------------------------------------
@dataclass
class Plan:
    task_names: List[str]
    #: per task: index into its option list (see ``_options``)
    chosen_option: List[int]
    #: per task: chosen gpu count
    gpu_counts: List[int]
    #: per task: concrete GPU ids occupied
    gpu_sets: List[List[int]]
    #: per task: scheduled start time (seconds from interval start)
    start_times: List[float]
    #: per task: runtime of the chosen strategy at solve time
    runtimes: List[float]
    makespan: float = 0.0
    #: solver diagnostics
    solver_status: str = "greedy"
    meta: Dict = field(default_factory=dict)

    def dependency_dict(self) -> Dict[int, List[int]]:
        """For each task index, the task indices that must finish first
        (tasks sharing >=1 GPU and scheduled earlier)."""
        deps: Dict[int, List[int]] = {i: [] for i in range(len(self.task_names))}
        for i in range(len(self.task_names)):
            for j in range(len(self.task_names)):
                if i == j:
                    continue
                if set(self.gpu_sets[i]) & set(self.gpu_sets[j]):
                    # j precedes i if it starts earlier (ties: lower index)
                    if (self.start_times[j], j) < (self.start_times[i], i):
                        deps[i].append(j)
        return deps

    def shift(self, seconds: float) -> None:
        """Advance the plan by one executed interval (reference
        milp.py:434-442).

        A task that was RUNNING during the shifted window has executed
        ``min(seconds - start, runtime)`` of its work, so its remaining
        runtime shrinks along with the clamped start — otherwise the kept
        incumbent shows phantom overlaps between a task whose start
        clamped to 0 and its successors (fuzz-caught in round 2; the
        reference avoids it only because it destructively decrements
        strategy runtimes elsewhere, executor.py:166-172)."""
        for i in range(len(self.task_names)):
            s, r = self.start_times[i], self.runtimes[i]
            executed = min(max(seconds - s, 0.0), r)
            self.start_times[i] = max(0.0, s - seconds)
            self.runtimes[i] = r - executed
        self.makespan = max(0.0, self.makespan - seconds)

    def restrict(self, names: List[str]) -> "Plan":
        """A copy of this plan covering only ``names`` (tasks that survive
        into the next interval), so hysteresis comparison stays meaningful
        after task retirement."""
        idx = [self.task_names.index(nm) for nm in names]
        return Plan(
            task_names=[self.task_names[i] for i in idx],
            chosen_option=[self.chosen_option[i] for i in idx],
            gpu_counts=[self.gpu_counts[i] for i in idx],
            gpu_sets=[list(self.gpu_sets[i]) for i in idx],
            start_times=[self.start_times[i] for i in idx],
            runtimes=[self.runtimes[i] for i in idx],
            makespan=self.makespan,
            solver_status=self.solver_status,
        )


# ---------------------------------------------------------------------------
# Option extraction
# ---------------------------------------------------------------------------
def _remaining_runtime(task, strat) -> float:
    """Remaining whole-job runtime for a strategy: per-batch time x batches
    left.  Non-destructive replacement for the reference's in-place
    ``strategies[g].runtime -=`` bookkeeping (executor.py:166-172)."""
    if strat.batch_time is not None:
        return float(strat.batch_time) * max(1, task.remaining_batches)
    if strat.runtime is None:
        return INFEASIBLE_RUNTIME
    return float(strat.runtime)


def _options(task) -> List[Tuple[int, float]]:
    """(gpu_count, remaining_runtime) options for a task, in sorted-g order.

    Prefer feasible cells; if none exist keep every cell with its sentinel
    runtime so the batch stays schedulable and the MILP merely avoids the
    task's cost (reference keeps 1e6-runtime sentinel strategies,
    PerformanceEvaluator.py:96-99).
    """
    cells = sorted(task.strategies.items())
    feasible = [
        (g, _remaining_runtime(task, s))
        for g, s in cells
        if s is not None and s.feasible and s.runtime is not None
    ]
    if feasible:
        return feasible
    return [(g, _remaining_runtime(task, s)) for g, s in cells]


# ---------------------------------------------------------------------------
# Greedy fallback: earliest-finish list scheduling
# ---------------------------------------------------------------------------
def _greedy_plan(task_list, n_gpus: int) -> Plan:
    opts = [_options(t) for t in task_list]
    free_at = np.zeros(n_gpus)  # per-GPU time when it becomes free
    order = sorted(
        range(len(task_list)),
        key=lambda i: -min(r for _, r in opts[i]),  # longest job first
    )
    chosen = [0] * len(task_list)
    gpu_sets: List[List[int]] = [[] for _ in task_list]
    starts = [0.0] * len(task_list)
    runtimes = [0.0] * len(task_list)
    for i in order:
        best = None  # (finish, start, opt_idx, gpus)
        for k, (g, r) in enumerate(opts[i]):
            if g > n_gpus:
                continue
            ids = np.argsort(free_at, kind="stable")[:g]
            start = float(free_at[ids].max())
            cand = (start + r, start, k, [int(x) for x in ids])
            if best is None or cand[0] < best[0]:
                best = cand
        if best is None:
            raise ValueError(
                f"Task {task_list[i].name} needs more GPUs than the node has."
            )
        finish, start, k, ids = best
        chosen[i] = k
        gpu_sets[i] = ids
        starts[i] = start
        runtimes[i] = opts[i][k][1]
        for g_id in ids:
            free_at[g_id] = finish
    return Plan(
        task_names=[t.name for t in task_list],
        chosen_option=chosen,
        gpu_counts=[opts[i][chosen[i]][0] for i in range(len(task_list))],
        gpu_sets=gpu_sets,
        start_times=starts,
        runtimes=runtimes,
        makespan=float(free_at.max()) if len(task_list) else 0.0,
        solver_status="greedy",
    )


# ---------------------------------------------------------------------------
# MILP formulation
# ---------------------------------------------------------------------------
def _milp_plan(
    task_list,
    n_gpus: int,
    timeout: float,
    incumbent_bound: Optional[float] = None,
) -> Optional[Plan]:
    """``incumbent_bound``: an upper bound on the makespan of any plan worth
    adopting (previous plan's remaining makespan minus the hysteresis
    margin).  scipy's HiGHS interface cannot take a warm-start solution
    (the reference warm-starts Gurobi, milp.py:322-327), so the bound is
    how the previous incumbent prunes the branch-and-bound tree; when no
    plan beats it the model is infeasible, which HiGHS usually proves
    quickly, and the caller keeps the shifted previous plan."""
    T = len(task_list)
    if T == 0:
        return Plan([], [], [], [], [], [], 0.0, "empty")
    opts = [_options(t) for t in task_list]

    # Horizon / big-M: everything sequential.
    horizon = sum(max(r for _, r in o) for o in opts) + 1.0
    M = horizon

    # ---- variable layout -------------------------------------------------
    # [x(t,k)...][occ(t,g)...][start(t)...][after(i,j) i<j ...][makespan]
    x_off: List[int] = []
    n = 0
    for o in opts:
        x_off.append(n)
        n += len(o)
    occ_off = n
    n += T * n_gpus
    start_off = n
    n += T
    pair_index: Dict[Tuple[int, int], int] = {}
    after_off = n
    for i in range(T):
        for j in range(i + 1, T):
            pair_index[(i, j)] = n
            n += 1
    mk = n
    n += 1

    integrality = np.zeros(n)
    lb = np.zeros(n)
    ub = np.full(n, np.inf)
    for t in range(T):
        for k in range(len(opts[t])):
            integrality[x_off[t] + k] = 1
            ub[x_off[t] + k] = 1
    integrality[occ_off : occ_off + T * n_gpus] = 1
    ub[occ_off : occ_off + T * n_gpus] = 1
    for (i, j), v in pair_index.items():
        integrality[v] = 1
        ub[v] = 1
    ub[start_off : start_off + T] = horizon
    ub[mk] = horizon
    if incumbent_bound is not None:
        ub[mk] = min(horizon, float(incumbent_bound))

    rows: List[Dict[int, float]] = []
    lo: List[float] = []
    hi: List[float] = []

    def add(row: Dict[int, float], lo_v: float, hi_v: float) -> None:
        rows.append(row)
        lo.append(lo_v)
        hi.append(hi_v)

    # one strategy per task
    for t in range(T):
        add({x_off[t] + k: 1.0 for k in range(len(opts[t]))}, 1.0, 1.0)

    # gpu-count consistency: sum_g occ = sum_k g_k x
    for t in range(T):
        row = {occ_off + t * n_gpus + g: 1.0 for g in range(n_gpus)}
        for k, (g_k, _) in enumerate(opts[t]):
            row[x_off[t] + k] = -float(g_k)
        add(row, 0.0, 0.0)

    # makespan >= start + runtime
    for t in range(T):
        row = {mk: 1.0, start_off + t: -1.0}
        for k, (_, r_k) in enumerate(opts[t]):
            row[x_off[t] + k] = -float(r_k)
        add(row, 0.0, np.inf)

    # pairwise no-overlap on shared GPUs
    for i in range(T):
        for j in range(i + 1, T):
            a = pair_index[(i, j)]
            for g in range(n_gpus):
                oi = occ_off + i * n_gpus + g
                oj = occ_off + j * n_gpus + g
                # after=1 -> j after i:  start_j - start_i - dur_i
                #   >= -M(3 - occ_i - occ_j - after)
                row = {
                    start_off + j: 1.0,
                    start_off + i: -1.0,
                    oi: -M,
                    oj: -M,
                    a: -M,
                }
                for k, (_, r_k) in enumerate(opts[i]):
                    row[x_off[i] + k] = -float(r_k)
                add(row, -3.0 * M, np.inf)
                # after=0 -> i after j:  start_i - start_j - dur_j
                #   >= -M(2 - occ_i - occ_j + after)
                row = {
                    start_off + i: 1.0,
                    start_off + j: -1.0,
                    oi: -M,
                    oj: -M,
                    a: M,
                }
                for k, (_, r_k) in enumerate(opts[j]):
                    row[x_off[j] + k] = -float(r_k)
                add(row, -2.0 * M, np.inf)

    data, indices, indptr = [], [], [0]
    for row in rows:
        for c, v in sorted(row.items()):
            indices.append(c)
            data.append(v)
        indptr.append(len(indices))
    A = csr_matrix((data, indices, indptr), shape=(len(rows), n))

    c = np.zeros(n)
    c[mk] = 1.0

    res = milp(
        c,
        constraints=LinearConstraint(A, np.array(lo), np.array(hi)),
        integrality=integrality,
        bounds=Bounds(lb, ub),
        options={"time_limit": float(timeout), "mip_rel_gap": 1e-4},
    )
    if res.x is None:
        return None

    xv = res.x
    chosen = []
    gpu_sets = []
    for t in range(T):
        ks = [xv[x_off[t] + k] for k in range(len(opts[t]))]
        chosen.append(int(np.argmax(ks)))
        gpu_sets.append(
            [g for g in range(n_gpus) if round(xv[occ_off + t * n_gpus + g]) == 1]
        )
    plan = Plan(
        task_names=[t.name for t in task_list],
        chosen_option=chosen,
        gpu_counts=[opts[t][chosen[t]][0] for t in range(T)],
        gpu_sets=gpu_sets,
        start_times=[float(xv[start_off + t]) for t in range(T)],
        runtimes=[float(opts[t][chosen[t]][1]) for t in range(T)],
        makespan=float(xv[mk]),
        solver_status="optimal" if res.status == 0 else f"status_{res.status}",
    )
    _repair_overlaps(plan)
    return plan


def _repair_overlaps(plan: Plan) -> None:
    """HiGHS honors constraints only to its ~1e-6 feasibility tolerance, so
    two tasks sharing a GPU can overlap by a microsecond in the raw
    solution.  Execution is dependency-ordered (engine events), so this only
    tidies the reported schedule: push each start to the latest end among
    earlier-started tasks sharing a GPU, in start order."""
    order = sorted(range(len(plan.task_names)),
                   key=lambda i: (plan.start_times[i], i))
    for pos, i in enumerate(order):
        floor = 0.0
        for j in order[:pos]:
            if set(plan.gpu_sets[i]) & set(plan.gpu_sets[j]):
                floor = max(floor, plan.start_times[j] + plan.runtimes[j])
        if plan.start_times[i] < floor:
            plan.start_times[i] = floor
    plan.makespan = max(
        [plan.start_times[i] + plan.runtimes[i]
         for i in range(len(plan.task_names))],
        default=0.0,
    )


# ---------------------------------------------------------------------------
# Public API
# ---------------------------------------------------------------------------
def solve(
    task_list,
    presolved: Optional[Plan] = None,
    interval: float = 1000.0,
    timeout: float = 500.0,
    n_gpus: Optional[int] = None,
    hysteresis: float = 500.0,
    threads: Optional[int] = None,  # accepted for API parity; HiGHS decides
) -> Plan:
    """Solve (or re-solve) the gang schedule.

    With ``presolved`` given and still covering the same task set, the new
    plan is adopted only if its makespan beats the saved plan's remaining
    makespan by more than ``interval + hysteresis`` (reference
    milp.py:363-381); otherwise the saved plan is shifted by ``interval``.

    Warm start: the previous plan's remaining makespan (minus the
    hysteresis margin) is fed to the MILP as an upper bound on the
    makespan variable — the scipy/HiGHS equivalent of the reference's
    Gurobi ``warmStart`` incumbent (milp.py:322-327): the tree is pruned
    to only plans worth adopting, and on timeout/infeasibility the
    *shifted previous plan* is kept instead of degrading to greedy.
    """
    if n_gpus is None:
        n_gpus = detect_gpu_count()
    names = [t.name for t in task_list]
    can_keep = presolved is not None and presolved.task_names == names
    covers = presolved is not None and set(presolved.task_names) >= set(names)
    bound = None
    if covers:
        bound = max(0.0, presolved.makespan - interval - hysteresis)
    plan = _milp_plan(task_list, n_gpus, timeout, incumbent_bound=bound)
    if plan is None:
        if covers:
            kept = presolved if can_keep else presolved.restrict(names)
            kept.shift(interval)
            kept.solver_status = "kept_incumbent"
            log.info(
                "no plan beats the previous one within %.0fs; keeping the "
                "shifted incumbent (makespan %.1f)", timeout, kept.makespan,
            )
            return kept
        log.warning(
            "MILP produced no incumbent within %.0fs; greedy fallback",
            timeout,
        )
        return _greedy_plan(task_list, n_gpus)

    if can_keep:
        if plan.makespan < presolved.makespan - interval - hysteresis:
            return plan
        kept = presolved
        kept.shift(interval)
        return kept
    return plan


def detect_gpu_count(default: int = 8) -> int:
    try:
        import torch

        if torch.cuda.is_available():
            return torch.cuda.device_count()
    except Exception:
        pass
    return default


def apply_plan(task_list, plan: Plan) -> None:
    """Write strategy selections back onto the tasks (explicit option
    mapping — no dict-insertion-order coupling, reference quirk
    milp.py:478-486)."""
    for t_idx, task in enumerate(task_list):
        g = plan.gpu_counts[t_idx]
        strat = task.strategies.get(g)
        if strat is None:
            raise KeyError(
                f"Plan chose {g} GPUs for task {task.name} but no strategy "
                "was profiled at that count."
            )
        task.select_strategy(strat)


def convert_into_comprehensible(task_list, plan: Plan):
    """API-parity shim (reference milp.py:448-513): returns
    (node_per_task, task_dependency_dict, start_time_per_task).  Single-node
    build: node is always 0."""
    apply_plan(task_list, plan)
    node_per_task = {t: 0 for t in task_list}
    deps_idx = plan.dependency_dict()
    task_dependency_dict = {
        task_list[i]: [task_list[j] for j in deps]
        for i, deps in deps_idx.items()
    }
    return node_per_task, task_dependency_dict, list(plan.start_times)
