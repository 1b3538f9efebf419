from .ddp import DDPExecutor
from .fsdp import FSDPExecutor
from .pipeline import PipelineExecutor
from .spilled import SpilledExecutor

__all__ = ["DDPExecutor", "FSDPExecutor", "PipelineExecutor", "SpilledExecutor"]
