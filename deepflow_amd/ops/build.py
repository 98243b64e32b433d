"""In-tree native build: libdfcpu.so (g++) and libdfgpu.so (hipcc, gfx950).

Built artifacts live next to this file so they travel with the repo snapshot
to GPU boxes (they are git-ignored but NOT gpurun-ignored). Rebuilds happen
only when a source file is newer than the artifact.
"""
from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
CSRC = OPS_DIR / "csrc"
CPU_LIB = OPS_DIR / "libdfcpu.so"
GPU_LIB = OPS_DIR / "libdfgpu.so"

CPU_SOURCES = ["dfcpu.cpp", "agent_core.cpp", "otlp_conv.cpp",
               "recv_pump.cpp"]
GPU_SOURCES = ["dfgpu.hip"]

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
GFX_ARCH = os.environ.get("DF_GFX_ARCH", "gfx950")


def _needs_build(lib: Path, sources) -> bool:
    if not lib.exists():
        return True
    lib_mtime = lib.stat().st_mtime
    deps = [CSRC / s for s in sources] + list(CSRC.glob("*.h")) + [Path(__file__)]
    return any(d.exists() and d.stat().st_mtime > lib_mtime for d in deps)


def _run(cmd):
    proc = subprocess.run(cmd, capture_output=True, text=True)
    if proc.returncode != 0:
        raise RuntimeError(
            f"build failed: {' '.join(cmd)}\n{proc.stdout}\n{proc.stderr}")


def build_cpu(force: bool = False) -> Path:
    srcs = [str(CSRC / s) for s in CPU_SOURCES if (CSRC / s).exists()]
    if force or _needs_build(CPU_LIB, CPU_SOURCES):
        _run(["g++", "-O3", "-std=c++17", "-shared", "-fPIC", "-fopenmp",
              "-pthread", "-ldl",
              *srcs, "-o", str(CPU_LIB)])
    return CPU_LIB


def build_gpu(force: bool = False) -> Path:
    srcs = [str(CSRC / s) for s in GPU_SOURCES if (CSRC / s).exists()]
    if not srcs:
        raise RuntimeError("no GPU sources present")
    if force or _needs_build(GPU_LIB, GPU_SOURCES):
        _run([HIPCC, f"--offload-arch={GFX_ARCH}", "-O3", "-std=c++17",
              "-shared", "-fPIC", *srcs, "-o", str(GPU_LIB)])
    return GPU_LIB


PROF_DIR = OPS_DIR.parent / "profiler"
PROF_LIB = PROF_DIR / "libdfprof.so"


def build_prof(force: bool = False) -> Path:
    """roctracer subscriber for the continuous GPU profiler."""
    src = PROF_DIR / "csrc" / "gpuprof.cpp"
    if force or not PROF_LIB.exists() or \
            src.stat().st_mtime > PROF_LIB.stat().st_mtime:
        # plain g++: a rocprofiler tool library must NOT embed HIP
        # runtime registration stubs (hipcc links them and the runtime
        # then aborts in rocprofiler_set_api_table when the tool loads)
        _run(["g++", "-O2", "-std=c++17", "-shared", "-fPIC",
              "-D__HIP_PLATFORM_AMD__", str(src),
              "-I/opt/rocm/include", "-L/opt/rocm/lib", "-lrocprofiler-sdk",
              "-Wl,-rpath,/opt/rocm/lib", "-o", str(PROF_LIB)])
    return PROF_LIB


def build_all(force: bool = False) -> None:
    build_prof(force)
    build_cpu(force)
    build_gpu(force)


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
    print(f"built: {CPU_LIB} {GPU_LIB}")
