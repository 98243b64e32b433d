from .gpu_profiler import GpuProfiler  # noqa: F401
