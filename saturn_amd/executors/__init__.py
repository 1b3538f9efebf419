from .ddp import DDPExecutor
from .fsdp import FSDPExecutor
from .pipeline import PipelineExecutor
from .megatron import MegatronExecutor
from .spilled import SpilledExecutor
from .ulysses import UlyssesExecutor
from .expert import ExpertParallelExecutor

__all__ = ["DDPExecutor", "FSDPExecutor", "PipelineExecutor", "SpilledExecutor", "MegatronExecutor", "UlyssesExecutor", "ExpertParallelExecutor"]
