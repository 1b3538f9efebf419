from .milp import (
    Plan,
    apply_plan,
    convert_into_comprehensible,
    detect_gpu_count,
    solve,
)

__all__ = [
    "Plan",
    "solve",
    "apply_plan",
    "convert_into_comprehensible",
    "detect_gpu_count",
]
