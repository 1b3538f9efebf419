"""Native RCCL comm engine wrapper.

``create_comm(rank, world)`` bootstraps a per-job communicator: rank 0
generates the ncclUniqueId and ships it to peers through the job's TCP
rendezvous (127.0.0.1, port pool keyed by task id — the same channel the
process group uses).  When torch.distributed is already initialized the id
rides its store; otherwise a FileStore under the job's save dir works.

The engine runs collectives on a dedicated high-priority HIP stream with
event fencing against torch's compute stream, so bucket all-reduces overlap
backward without torch's ProcessGroup in the path.
"""

from __future__ import annotations

import importlib
from typing import Optional

_ext = None
_err: Optional[str] = None


def _load():
    global _ext, _err
    if _ext is not None or _err is not None:
        return _ext
    try:
        _ext = importlib.import_module("saturn_amd._comm")
    except Exception as e:  # noqa: BLE001
        _err = f"{type(e).__name__}: {e}"
        _ext = None
    return _ext


def has_native_comm() -> bool:
    return _load() is not None


def require_native_comm():
    ext = _load()
    if ext is None:
        raise RuntimeError(
            f"saturn_amd._comm (native RCCL engine) not built: {_err}"
        )
    return ext


class _Comm:
    """Thin keep-alive wrapper over the native RcclComm.

    Collectives run on the engine's own HIP stream; a caller that drops a
    tensor right after the call could otherwise have the caching allocator
    reuse its memory while the collective still reads it (the hazard
    torch's ProcessGroupNCCL solves with recordStream).  Holding a Python
    reference until ``join()`` is the simple, allocator-agnostic fix.
    """

    def __init__(self, inner):
        self._c = inner
        self._pending = []

    @property
    def rank(self):
        return self._c.rank

    @property
    def world(self):
        return self._c.world

    def all_reduce(self, t, average=True):
        self._pending.append(t)
        self._c.all_reduce(t, average)

    def broadcast(self, t, root):
        self._pending.append(t)
        self._c.broadcast(t, root)

    def all_gather(self, out, inp):
        self._pending.extend((out, inp))
        self._c.all_gather(out, inp)

    def reduce_scatter(self, out, inp, average=False):
        self._pending.extend((out, inp))
        self._c.reduce_scatter(out, inp, average)

    def send(self, t, peer):
        self._pending.append(t)
        self._c.send(t, peer)

    def recv(self, t, peer):
        self._pending.append(t)
        self._c.recv(t, peer)

    def join(self):
        # compute stream now waits the comm stream: anything enqueued so
        # far is ordered before future compute, so the references can drop
        self._c.join()
        self._pending.clear()

    def synchronize(self):
        self._c.synchronize()
        self._pending.clear()


def create_comm(rank: int, world: int):
    """Create a communicator for the current process group's gang.  Uses
    torch.distributed (already initialized by the executor launch) only to
    move the 128-byte unique id."""
    ext = require_native_comm()
    import torch.distributed as dist

    if world == 1:
        return _Comm(ext.RcclComm(ext.get_unique_id(), 0, 1))
    assert dist.is_initialized(), "init_process_group first (id exchange)"
    obj = [ext.get_unique_id() if rank == 0 else None]
    dist.broadcast_object_list(obj, src=0)
    return _Comm(ext.RcclComm(obj[0], rank, world))
