"""saturn-amd: MI355X-native multi-large-model training orchestration.

A brand-new framework with the capabilities of knagrecha/saturn (reference
API surface: ``saturn/__init__.py``, ``saturn/core/representations``,
``saturn/library``, ``saturn/trial_runner``), built for a single 8xMI355X
node: PyTorch-ROCm + hand-written HIP/CDNA4 kernels for the hot ops, RCCL
over xGMI for collectives, scipy/HiGHS for the MILP gang scheduler, plain
multiprocessing for the control plane.

Public API (mirrors the reference's five entry points):

    Task, HParams            -- job specification
    BaseTechnique, Strategy  -- the UDP contract + its solver-facing tuple
    library.register/...     -- dill-backed technique library
    trial_runner.search      -- empirical profiling of (task, g, technique)
    orchestrate              -- interval-based introspective execution
"""

from saturn_amd.core import (
    BaseTechnique,
    HParams,
    Strategy,
    Task,
    Techniques,
)
from saturn_amd.library import deregister, register, retrieve
from saturn_amd.orchestrator import orchestrate
from saturn_amd.solver import Plan, solve
from saturn_amd.trial_runner import search

__version__ = "0.1.0"

__all__ = [
    "Task",
    "HParams",
    "Strategy",
    "Techniques",
    "BaseTechnique",
    "register",
    "deregister",
    "retrieve",
    "search",
    "orchestrate",
    "solve",
    "Plan",
]
