from .task import HParams, Task
from .strategy import INFEASIBLE_RUNTIME, Strategy, Techniques
from .technique import BaseTechnique

__all__ = [
    "HParams",
    "Task",
    "Strategy",
    "Techniques",
    "BaseTechnique",
    "INFEASIBLE_RUNTIME",
]
