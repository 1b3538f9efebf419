"""Strategy: a (technique, GPU count, tuned parameters, runtime) tuple.

Parity with reference ``saturn/core/representations/Strategy.py:25-76``
(Techniques enum + Strategy class), minus the Ray remoting helpers — the
MI355X build's control plane is plain multiprocessing.
"""

from __future__ import annotations

from enum import Enum
from typing import Any, Dict, Optional

#: Sentinel runtime for a (task, gpu-count) cell no technique could run
#: (reference PerformanceEvaluator.py:99 uses 1e6; the MILP's big-M must
#: stay well above this).
INFEASIBLE_RUNTIME = 1.0e6


class Techniques(Enum):
    """Built-in technique classes (reference Strategy.py:25-34 declares
    SPILLED/PIPELINE/FSDP/MEGATRON; the library is user-extensible beyond
    these)."""

    SPILLED = 1
    PIPELINE = 2
    FSDP = 3
    MEGATRON = 4
    DDP = 5


class Strategy:
    """A concrete execution choice for one task.

    Parameters
    ----------
    executor : the BaseTechnique subclass to run with (None = infeasible cell).
    gpu_apportionment : number of GPUs.
    parameters : tuned executor parameters from ``search()``
        (e.g. ``{"bucket_mb": 128}``).  The reference's DDP returned None here
        and was therefore never selectable (reference DDP.py:71-72 vs
        PerformanceEvaluator.py:110) — our executors always return a dict on
        success.
    runtime : estimated whole-job runtime in seconds (per-batch trial time x
        total batches).
    """

    def __init__(
        self,
        executor,
        gpu_apportionment: int,
        parameters: Optional[Dict[str, Any]] = None,
        runtime: Optional[float] = None,
        batch_time: Optional[float] = None,
    ) -> None:
        if not isinstance(gpu_apportionment, int) or gpu_apportionment <= 0:
            raise ValueError("GPU apportionment must be an integer > 0.")
        self.executor = executor
        self.gpu_apportionment = gpu_apportionment
        self.parameters = parameters
        self.runtime = runtime
        #: measured seconds per batch (the primitive quantity; ``runtime`` is
        #: batch_time x total_batches at profile time).  Keeping it here lets
        #: the solver compute *remaining* runtime without destructively
        #: decrementing Strategy.runtime the way the reference does
        #: (executor.py:166-172).
        self.batch_time = batch_time

    @property
    def feasible(self) -> bool:
        return self.executor is not None and self.parameters is not None

    def __repr__(self) -> str:
        name = getattr(self.executor, "name", self.executor)
        return (
            f"Strategy({name}, {self.gpu_apportionment}G, "
            f"params={self.parameters}, {self.runtime}s)"
        )
