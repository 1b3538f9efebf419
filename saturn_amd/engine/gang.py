"""Gang execution engine: dependency-ordered task launches on GPU gangs.

MI355X-native replacement for the reference's Ray actor stack
(``saturn/executor/executor.py:24-129``: DependencyHolder / LauncherActor /
ExecutorActor) and its patched ``mp.spawn``
(``saturn/core/executors/multiprocessing/my_multiprocessing.py``):

- GPU gangs are reserved by setting ``HIP_VISIBLE_DEVICES`` for one launcher
  subprocess per task (the reference used Ray's ``num_gpus`` actor option,
  executor.py:59-62);
- the MILP's ordering is enforced with one ``threading.Event`` per task in
  the parent (the reference used an asyncio-event Ray actor,
  executor.py:24-47);
- tasks and executors are shipped to the launcher subprocess with dill
  (closures in ``get_model``/``get_dataloader`` survive the boundary, same
  reason the reference stores UDPs with dill);
- child failures are re-raised in the parent with the original traceback
  (replaces ``processify.py:21-60`` and the my_multiprocessing join fix).
"""

from __future__ import annotations

import logging
import multiprocessing as mp
import os
import threading
import traceback
from timeit import default_timer as timer
from typing import Dict, List, Optional, Sequence

import dill

log = logging.getLogger(__name__)

_CTX = mp.get_context("spawn")


# ---------------------------------------------------------------------------
# Subprocess plumbing
# ---------------------------------------------------------------------------
def _child_entry(payload: bytes, env: Dict[str, str], out_q) -> None:
    """Runs in the launcher subprocess.  Sets GPU visibility BEFORE any HIP
    initialization, then executes the payload function."""
    try:
        os.environ.update(env)
        fn, args, kwargs = dill.loads(payload)
        result = fn(*args, **kwargs)
        out_q.put(("ok", dill.dumps(result)))
    except BaseException:
        out_q.put(("err", traceback.format_exc()))
        raise SystemExit(1)


def call_in_subprocess(
    fn,
    *args,
    env: Optional[Dict[str, str]] = None,
    timeout: Optional[float] = None,
    **kwargs,
):
    """Run ``fn(*args, **kwargs)`` in a fresh spawn-context process with
    ``env`` merged into its environment; return its result or re-raise the
    child error here with the original traceback."""
    out_q = _CTX.Queue()
    payload = dill.dumps((fn, args, kwargs), recurse=True)
    p = _CTX.Process(
        target=_child_entry, args=(payload, env or {}, out_q), daemon=False
    )
    p.start()
    # Read the result BEFORE join: a large payload can deadlock join()
    # against the queue's feeder thread otherwise.
    msg = None
    try:
        msg = out_q.get(timeout=timeout)
    except Exception:
        pass
    p.join(30 if msg is not None else 1)
    if p.is_alive():
        p.terminate()
        p.join(10)
        if msg is None:
            raise TimeoutError(f"subprocess for {fn} exceeded {timeout}s")
    if msg is None:
        raise RuntimeError(
            f"subprocess for {getattr(fn, '__name__', fn)} exited with "
            f"code {p.exitcode} and no result"
        )
    status, data = msg
    if status == "err":
        raise RuntimeError(
            f"subprocess for {getattr(fn, '__name__', fn)} failed:\n"
            f"--- child traceback ---\n{data}"
        )
    return dill.loads(data)


def run_in_subprocess(
    fn,
    *args,
    env: Optional[Dict[str, str]] = None,
    timeout: Optional[float] = None,
    **kwargs,
):
    """``call_in_subprocess`` with the result discarded."""
    call_in_subprocess(fn, *args, env=env, timeout=timeout, **kwargs)


# ---------------------------------------------------------------------------
# Gang execution for one interval
# ---------------------------------------------------------------------------
def _launch_task(task, gpu_ids: Sequence[int], tid: int, batch_count: int) -> None:
    """Launcher-subprocess body: run the task's selected executor on its
    gang.  Executors see logical GPUs [0..g-1]; HIP_VISIBLE_DEVICES has
    already mapped them onto the gang's physical devices."""
    executor = task.selected_strategy.executor
    executor.execute(task, list(range(len(gpu_ids))), tid, batch_count)


def execute(
    relevant_tasks: List,
    batches_to_run: List[int],
    interval: float,
    plan,
    task_dependency_dict: Optional[Dict] = None,
    launch_timeout: Optional[float] = None,
    raise_on_failure: bool = True,
    skip_ckpt_names: Optional[set] = None,
) -> List:
    """Execute one interval's tasks with MILP-ordered gang placement.

    Parity with reference ``executor.execute`` (executor.py:88-129): every
    relevant task is launched on its plan-assigned GPU set once all earlier
    tasks sharing any of its GPUs have finished; afterwards the parent
    advances each task's dataloader cursor.

    Returns the list of tasks that FAILED this interval.  With
    ``raise_on_failure=False`` the orchestrator can keep the rest of the
    batch running and retry failures from their last checkpoint (elastic
    recovery the reference lacks — a crash there kills the whole batch,
    SURVEY §5.3).
    """
    n = len(relevant_tasks)
    if n == 0:
        return []
    done_events = [threading.Event() for _ in range(n)]
    errors: List[Optional[str]] = [None] * n

    # Map plan indices -> relevant indices (the plan covers the full task
    # list; only tasks starting inside this interval run now).
    plan_idx = {nm: i for i, nm in enumerate(plan.task_names)}
    deps_by_plan = plan.dependency_dict()
    relevant_names = {t.name: i for i, t in enumerate(relevant_tasks)}

    def dep_indices(task) -> List[int]:
        out = []
        for j in deps_by_plan.get(plan_idx[task.name], []):
            nm = plan.task_names[j]
            if nm in relevant_names:
                out.append(relevant_names[nm])
        return out

    start = timer()

    def runner(r_idx: int) -> None:
        task = relevant_tasks[r_idx]
        try:
            for d in dep_indices(task):
                done_events[d].wait()
                if errors[d] is not None:
                    raise RuntimeError(
                        f"dependency task {relevant_tasks[d].name} failed"
                    )
            gpus = plan.gpu_sets[plan_idx[task.name]]
            visible = ",".join(str(g) for g in gpus)
            log.info(
                "launching %s on GPUs [%s] for %d batches",
                task.name,
                visible,
                batches_to_run[r_idx],
            )
            # Per-gang host-CPU reservation (reference executor.py:107
            # reserved num_cpus = g*3/4 per gang via Ray): cap the BLAS/
            # OMP thread pools to this gang's proportional share so
            # concurrent gangs don't oversubscribe host cores during
            # dataloading/checkpointing.
            n_vis = max(1, len(gpus))
            total_gpus = max(n_vis, len({g for s in plan.gpu_sets for g in s}))
            cpu_share = max(
                1, (os.cpu_count() or 8) * 3 // 4 * n_vis // total_gpus
            )
            env = {
                "HIP_VISIBLE_DEVICES": visible,
                "CUDA_VISIBLE_DEVICES": visible,
                "SATURN_TASK_ID": str(plan_idx[task.name]),
                "OMP_NUM_THREADS": str(cpu_share),
                "MKL_NUM_THREADS": str(cpu_share),
            }
            if skip_ckpt_names and task.name in skip_ckpt_names:
                # task completes inside this interval: no resume will ever
                # read its checkpoint, so skip the (potentially tens of
                # GB) write — disk hygiene for full-scale batches
                env["SATURN_SKIP_CKPT"] = "1" 
            run_in_subprocess(
                _launch_task,
                task,
                gpus,
                plan_idx[task.name],
                batches_to_run[r_idx],
                env=env,
                timeout=launch_timeout,
            )
            # Parent-side bookkeeping (child mutations don't propagate).
            task.reconfigure(batches_to_run[r_idx])
            log.info("task %s finished its interval quota", task.name)
        except BaseException:
            errors[r_idx] = traceback.format_exc()
        finally:
            done_events[r_idx].set()

    threads = [
        threading.Thread(target=runner, args=(i,), daemon=True) for i in range(n)
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join()

    failed_idx = [i for i, e in enumerate(errors) if e is not None]
    if failed_idx:
        details = "\n".join(
            f"--- {relevant_tasks[i].name} ---\n{errors[i]}" for i in failed_idx
        )
        if raise_on_failure:
            raise RuntimeError(
                f"{len(failed_idx)} task(s) failed this interval:\n{details}"
            )
        log.warning(
            "%d task(s) failed this interval (will retry from checkpoint):\n%s",
            len(failed_idx),
            details,
        )

    elapsed = timer() - start
    log.info("interval done: intended %.1fs actual %.1fs", interval, elapsed)
    return [relevant_tasks[i] for i in failed_idx]
