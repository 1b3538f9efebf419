from .ddp import DDPExecutor

__all__ = ["DDPExecutor"]
