"""Rendezvous port pool for per-job process groups.

Concurrent gang-scheduled jobs each create their own RCCL communicator;
disjoint rendezvous ports keyed by task id keep them isolated (the
reference used ``MASTER_PORT = 12000 + tid``, DDP.py:31).  The address is
always 127.0.0.1 — single-node build, and container hostnames may not
resolve.
"""

from __future__ import annotations

import os
from typing import Dict


def port_for(tid: int) -> int:
    base = int(os.environ.get("SATURN_PORT_BASE", "23100"))
    return base + (tid % 4000)


def rendezvous_env(tid: int, rank: int, world_size: int) -> Dict[str, str]:
    return {
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(port_for(tid)),
        "RANK": str(rank),
        "LOCAL_RANK": str(rank),
        "WORLD_SIZE": str(world_size),
    }
