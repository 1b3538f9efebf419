"""Per-executor gang launch: one process per GPU, rendezvous over 127.0.0.1.

Replaces the reference's patched ``mp.spawn``
(``saturn/core/executors/multiprocessing/my_multiprocessing.py``) and the
per-executor ``dist.init_process_group("nccl", ...)`` boilerplate
(DDP.py:28-34, FSDP.py:44-50).  Worker processes are spawn-context, get
their rank/world/rendezvous via env, and ship rank-0's return value back to
the launcher; any worker failure kills the gang and re-raises with the child
traceback.
"""

from __future__ import annotations

import multiprocessing as mp
import os
import traceback
from typing import Any, Dict, List, Optional

import dill

_CTX = mp.get_context("spawn")


def _worker(payload: bytes, rank: int, world: int, env: Dict[str, str], q) -> None:
    try:
        os.environ.update(env)
        fn, args, kwargs = dill.loads(payload)
        result = fn(rank, world, *args, **kwargs)
        if rank == 0:
            q.put(("ok", dill.dumps(result)))
    except BaseException:
        q.put(("err", f"[rank {rank}]\n" + traceback.format_exc()))
        raise SystemExit(1)


def gang_spawn(
    fn,
    world_size: int,
    tid: int,
    *args: Any,
    timeout: Optional[float] = None,
    extra_env: Optional[Dict[str, str]] = None,
    **kwargs: Any,
):
    """Spawn ``world_size`` workers running ``fn(rank, world, *args)``.

    Each worker gets MASTER_ADDR/PORT (port pool keyed by ``tid``), RANK,
    LOCAL_RANK, WORLD_SIZE in its environment.  Returns rank-0's return
    value.
    """
    from saturn_amd.utils.ports import rendezvous_env

    q = _CTX.Queue()
    payload = dill.dumps((fn, args, kwargs), recurse=True)
    procs: List[mp.Process] = []
    for rank in range(world_size):
        env = rendezvous_env(tid, rank, world_size)
        if extra_env:
            env.update(extra_env)
        p = _CTX.Process(
            target=_worker, args=(payload, rank, world_size, env, q), daemon=False
        )
        p.start()
        procs.append(p)

    result = None
    err = None
    try:
        status, data = q.get(timeout=timeout)
        if status == "ok":
            result = dill.loads(data)
        else:
            err = data
    except Exception:
        err = "gang produced no result (timeout or silent death)"

    if err is not None:
        for p in procs:
            if p.is_alive():
                p.terminate()
    for p in procs:
        p.join(30)
        if p.is_alive():
            p.kill()
            p.join(5)
    if err is None:
        bad = [p.exitcode for p in procs if p.exitcode not in (0, None)]
        if bad:
            err = f"worker exit codes {bad}"
    if err is not None:
        raise RuntimeError(f"gang of {world_size} failed:\n{err}")
    return result


def init_process_group(rank: int, world: int) -> str:
    """Initialize torch.distributed from the env set by ``gang_spawn``.

    Backend: ``nccl`` (= RCCL on ROCm) when a GPU is visible, else ``gloo``
    so the whole distributed path is testable on CPU.  Returns the backend
    chosen.
    """
    import torch
    import torch.distributed as dist

    backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(backend=backend, rank=rank, world_size=world)
    if backend == "nccl":
        torch.cuda.set_device(rank)
    return backend


def destroy_process_group() -> None:
    import torch.distributed as dist

    if dist.is_initialized():
        dist.destroy_process_group()
