"""BaseTechnique: the user-defined-parallelism (UDP) contract.

Parity with reference ``saturn/core/executors/Technique.py:24-45``: a
technique is a class with two static methods —

- ``execute(task, gpus, tid, batch_count)``: run ``batch_count`` batches of
  the task on the given logical GPU list, checkpoint, and return;
- ``search(task, gpus, tid)``: autotune technique parameters on the given
  GPUs and return ``(params_dict, per_batch_seconds)``; ``(None, t)`` marks
  the cell infeasible (e.g. OOM).

Note the search contract here is *per-batch* time: the trial runner
multiplies by ``task.total_batches`` (reference PerformanceEvaluator.py:26).
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Any, Dict, List, Optional, Tuple


class BaseTechnique(ABC):
    """Subclass, implement the two methods, then ``library.register`` it."""

    #: Human-readable name used in logs and plan dumps.
    name = "BaseTechnique (override when extending)"

    @staticmethod
    @abstractmethod
    def execute(task, gpus: List[int], tid: int, batch_count: int) -> None:
        """Train ``task`` for ``batch_count`` batches on ``gpus`` (logical
        indices within the gang; the engine sets HIP_VISIBLE_DEVICES)."""

    @staticmethod
    @abstractmethod
    def search(
        task, gpus: List[int], tid: int
    ) -> Tuple[Optional[Dict[str, Any]], float]:
        """Autotune on ``gpus``; return (params, seconds_per_batch) or
        (None, inf) if this cell cannot run."""
