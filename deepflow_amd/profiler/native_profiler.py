"""Native continuous GPU profiler: our own roctracer subscriber.

Replaces the round-1 kineto wrapper: libdfprof.so
registers an in-process roctracer activity pool; a capture window is
roctracer_enable/disable (microseconds, vs ~0.7 s of kineto session
setup), and kernel dispatch records aggregate natively into a
(kernel -> count/total/max) table drained by dfp_drain. Folded stacks
land in the same profile store as the eBPF OnCPU profiler, so one flame
API serves CPU and GPU (BASELINE config #5).
"""
from __future__ import annotations

import contextlib
import ctypes as ct
import struct
import time
from pathlib import Path
from typing import Dict, List, Optional, Tuple

from ..wire.metric import PROFILE_EVENT_EBPF_HBM_IN_USE

PROF_DIR = Path(__file__).resolve().parent
_lib: Optional[ct.CDLL] = None


def lib() -> ct.CDLL:
    global _lib
    if _lib is None:
        path = PROF_DIR / "libdfprof.so"
        if not path.exists():
            from ..ops import build
            build.build_prof()
        h = ct.CDLL(str(path))
        h.dfp_register.restype = ct.c_int
        h.dfp_ready.restype = ct.c_int
        h.dfp_start.restype = ct.c_int
        h.dfp_stop.restype = ct.c_int
        h.dfp_drain.restype = ct.c_uint64
        h.dfp_drain.argtypes = [ct.c_void_p, ct.c_uint64]
        h.dfp_record_count.restype = ct.c_uint64
        _lib = h
    return _lib


def ensure_early() -> bool:
    """Arrange tool registration. MUST run before the first HIP runtime
    touch (before torch initializes the GPU): rocprofiler loads the
    libraries named in ROCP_TOOL_LIBRARIES at runtime init and calls
    their exported rocprofiler_configure — we add libdfprof.so there and
    pre-load it so the python bindings share the same instance. Call at
    process start (bench.py honors DF_GPU_PROF=1; server main() always
    does)."""
    import os
    path = PROF_DIR / "libdfprof.so"
    if not path.exists():
        from ..ops import build
        build.build_prof()
    prev = os.environ.get("ROCP_TOOL_LIBRARIES", "")
    if str(path) not in prev:
        os.environ["ROCP_TOOL_LIBRARIES"] = \
            f"{path}:{prev}" if prev else str(path)
    lib()
    return True


def available() -> bool:
    """Registered early enough for capture windows to work?"""
    try:
        return lib().dfp_ready() == 2
    except OSError:
        return False


def drain() -> List[Tuple[str, int, int, int]]:
    """-> [(kernel_name, count, total_ns, max_ns)], resetting the table."""
    h = lib()
    need = int(h.dfp_drain(None, 0))
    if need == 0:
        return []
    buf = ct.create_string_buffer(need)
    got = int(h.dfp_drain(ct.addressof(buf), need))
    out = []
    pos = 0
    raw = buf.raw[:got]
    while pos + 4 <= got:
        (nl,) = struct.unpack_from("<I", raw, pos)
        pos += 4
        name = raw[pos:pos + nl].decode("utf-8", "replace")
        pos += nl
        count, total, mx = struct.unpack_from("<QQQ", raw, pos)
        pos += 24
        out.append((name, count, total, mx))
    return out


class NativeGpuProfiler:
    """Window-based capture into the profile pipeline."""

    EVENT_TYPE_ON_GPU = PROFILE_EVENT_EBPF_HBM_IN_USE + 1  # OnGpu

    def __init__(self, pipeline, process_name: str = "deepflow-gpu"):
        self.pipe = pipeline
        self.process_name = process_name
        self.captures = 0
        self.window_overhead_ns = 0

    @contextlib.contextmanager
    def capture(self):
        h = lib()
        t0 = time.perf_counter_ns()
        rc = h.dfp_start()
        t1 = time.perf_counter_ns()
        if rc == -10:
            raise RuntimeError(
                "dfprof not registered before HIP init — call "
                "native_profiler.ensure_early() at process start "
                "(DF_GPU_PROF=1 for bench.py)")
        if rc != 0:
            raise RuntimeError(f"dfp_start failed rc={rc}")
        try:
            yield self
        finally:
            t2 = time.perf_counter_ns()
            h.dfp_stop()
            self._harvest()
            t3 = time.perf_counter_ns()
            self.window_overhead_ns += (t1 - t0) + (t3 - t2)
            self.captures += 1

    def _harvest(self) -> None:
        ts = int(time.time() * 1_000_000)
        common = dict(event_type=self.EVENT_TYPE_ON_GPU, pid=0, tid=0,
                      pod_id=0, process_name=self.process_name,
                      app_service="gpu", profile_language_type="roctracer")
        for name, count, total_ns, _mx in drain():
            stack = f"gpu;{name.replace(';', '_')}".encode()
            # value: total device-side microseconds
            self.pipe._add(ts, stack, max(total_ns // 1000, 1), common)


class ContinuousNativeProfiler(NativeGpuProfiler):
    """1 window per `interval_s`: because a window costs microseconds of
    host time (enable/disable + drain), the duty cycle can be far denser
    than kineto's ~0.7 s windows allowed; capture-window GPU overhead is
    the roctracer timestamping on dispatches inside the window only."""

    def __init__(self, pipeline, process_name: str = "deepflow-gpu",
                 interval_s: float = 60.0, window_s: float = 1.0):
        super().__init__(pipeline, process_name)
        self.interval_s = interval_s
        self.window_s = window_s
        self._win_until = 0.0
        self._next_win = 0.0
        self._in_window = False

    def step(self) -> None:
        """Call once per pipeline step; opens/closes windows on time."""
        now = time.monotonic()
        h = lib()
        if self._in_window and now >= self._win_until:
            h.dfp_stop()
            self._harvest()
            self._in_window = False
            self.captures += 1
            self._next_win = now + self.interval_s
        elif not self._in_window and now >= self._next_win:
            h.dfp_start()
            self._win_until = now + self.window_s
            self._in_window = True

    def close(self) -> None:
        if self._in_window:
            lib().dfp_stop()
            self._harvest()
            self._in_window = False
