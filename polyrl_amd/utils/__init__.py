from .profiling import (GPUMemoryLogger, annotate, log_gpu_memory,
                        roctx_range, step_profiler)

__all__ = ["roctx_range", "annotate", "GPUMemoryLogger", "log_gpu_memory",
           "step_profiler"]
