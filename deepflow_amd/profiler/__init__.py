from .gpu_profiler import GpuProfiler  # noqa: F401
from .native_profiler import (NativeGpuProfiler,  # noqa: F401
                              ContinuousNativeProfiler)
