from .ddp import DDPExecutor
from .fsdp import FSDPExecutor
from .pipeline import PipelineExecutor
from .megatron import MegatronExecutor
from .spilled import SpilledExecutor

__all__ = ["DDPExecutor", "FSDPExecutor", "PipelineExecutor", "SpilledExecutor", "MegatronExecutor"]
