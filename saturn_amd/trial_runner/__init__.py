from .evaluator import search
from .profiler import device_memory_stats, rocm_smi_sample, rocprof_stats

__all__ = ["search", "rocprof_stats", "rocm_smi_sample", "device_memory_stats"]
