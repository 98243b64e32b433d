"""ctypes bindings for libdfcpu.so (host) and libdfgpu.so (HIP/gfx950).

GPU policy: on a machine with a visible GPU the HIP library is REQUIRED —
ops fail loudly rather than falling back to eager PyTorch, so a passing GPU
test run always means the native kernels executed.
"""
from __future__ import annotations

import ctypes as ct
from pathlib import Path
from typing import Optional

OPS_DIR = Path(__file__).resolve().parent

_cpu_lib: Optional[ct.CDLL] = None
_gpu_lib: Optional[ct.CDLL] = None


class SpanCfgC(ct.Structure):
    _fields_ = [
        ("seed", ct.c_uint64),
        ("base_time_ns", ct.c_uint64),
        ("dt_ns", ct.c_uint64),
        ("n_agents", ct.c_uint32),
        ("n_ips", ct.c_uint32),
        ("n_epcs", ct.c_uint32),
        ("n_services", ct.c_uint32),
        ("n_resources", ct.c_uint32),
        ("tag_cardinality", ct.c_uint32),
        ("n_attrs", ct.c_uint32),
        ("err_rate_pct", ct.c_uint32),
        ("ip6_rate_pct", ct.c_uint32),
    ]


def span_cfg_c(cfg) -> SpanCfgC:
    """Convert gen.spans.SpanGenConfig -> C struct."""
    return SpanCfgC(
        seed=cfg.seed, base_time_ns=cfg.base_time_ns, dt_ns=cfg.dt_ns,
        n_agents=cfg.n_agents, n_ips=cfg.n_ips, n_epcs=cfg.n_epcs,
        n_services=cfg.n_services, n_resources=cfg.n_resources,
        tag_cardinality=cfg.tag_cardinality, n_attrs=cfg.n_attrs,
        err_rate_pct=cfg.err_rate_pct,
        ip6_rate_pct=getattr(cfg, "ip6_rate_pct", 0),
    )


def cpu() -> ct.CDLL:
    global _cpu_lib
    if _cpu_lib is None:
        path = OPS_DIR / "libdfcpu.so"
        if not path.exists():
            from . import build
            build.build_cpu()
        lib = ct.CDLL(str(path))
        lib.df_gen_spans.restype = ct.c_uint64
        lib.df_gen_spans.argtypes = [ct.POINTER(SpanCfgC), ct.c_uint64,
                                     ct.c_uint64, ct.c_void_p, ct.c_uint64]
        lib.df_gen_spans_indexed.restype = ct.c_uint64
        lib.df_gen_spans_indexed.argtypes = [
            ct.POINTER(SpanCfgC), ct.c_uint64, ct.c_uint64, ct.c_void_p,
            ct.c_uint64, ct.c_void_p, ct.c_void_p]
        lib.df_gen_spans_parallel.restype = ct.c_uint64
        lib.df_gen_spans_parallel.argtypes = [
            ct.POINTER(SpanCfgC), ct.c_uint64, ct.c_uint64, ct.c_void_p,
            ct.c_uint64, ct.c_void_p, ct.c_void_p]
        lib.df_otlp_to_l7.restype = ct.c_int64
        lib.df_otlp_to_l7.argtypes = [ct.c_void_p, ct.c_uint64,
                                      ct.c_void_p, ct.c_uint64]
        lib.df_route_spans.restype = ct.c_uint64
        lib.df_route_spans.argtypes = [ct.c_void_p, ct.c_uint64, ct.c_void_p,
                                       ct.c_void_p, ct.c_uint64, ct.c_uint32,
                                       ct.c_uint32, ct.c_void_p]
        lib.df_scan_offsets.restype = ct.c_uint64
        lib.df_scan_offsets.argtypes = [ct.c_void_p, ct.c_uint64, ct.c_void_p,
                                        ct.c_void_p, ct.c_uint64]
        lib.df_zstd_decompress.restype = ct.c_int64
        lib.df_zstd_decompress.argtypes = [ct.c_void_p, ct.c_uint64,
                                           ct.c_void_p, ct.c_uint64]
        lib.df_pump_start.restype = ct.c_void_p
        lib.df_pump_start.argtypes = [ct.c_int, ct.c_void_p, ct.c_uint64,
                                      ct.c_int]
        lib.df_pump_head.restype = ct.c_uint64
        lib.df_pump_head.argtypes = [ct.c_void_p]
        lib.df_pump_tail.restype = ct.c_uint64
        lib.df_pump_tail.argtypes = [ct.c_void_p]
        lib.df_pump_set_tail.restype = None
        lib.df_pump_set_tail.argtypes = [ct.c_void_p, ct.c_uint64]
        lib.df_pump_done.restype = ct.c_int
        lib.df_pump_done.argtypes = [ct.c_void_p]
        lib.df_pump_stats.restype = None
        lib.df_pump_stats.argtypes = [ct.c_void_p] + [ct.c_void_p] * 4
        lib.df_tcp_blast.restype = ct.c_int64
        lib.df_tcp_blast.argtypes = [ct.c_int, ct.c_void_p, ct.c_uint64,
                                     ct.c_uint32]
        lib.df_pump_free.restype = None
        lib.df_pump_free.argtypes = [ct.c_void_p]
        lib.df_zstd_compress.restype = ct.c_int64
        lib.df_zstd_compress.argtypes = [ct.c_void_p, ct.c_uint64,
                                         ct.c_void_p, ct.c_uint64, ct.c_int]
        _cpu_lib = lib
    return _cpu_lib


def _decl_gpu(lib: ct.CDLL) -> None:
    u64, u32, p = ct.c_uint64, ct.c_uint32, ct.c_void_p
    lib.df_gpu_ready.restype = ct.c_int
    lib.df_decode_l7.restype = ct.c_int
    lib.df_decode_l7.argtypes = [p, p, p, u32, p, p, p, p, p, p, u64, u64, u64, u64]
    lib.df_decode_l4.restype = ct.c_int
    lib.df_decode_l4.argtypes = [p, p, p, u32, p, p, p, p, u64, u64, u64, u64]
    lib.df_rollup_l4.restype = ct.c_int
    lib.df_rollup_l4.argtypes = [p, p, p, u64, u64, u32, u64, ct.c_char_p,
                                 u32, p, p, u64]
    lib.df_rollup_l7.restype = ct.c_int
    lib.df_rollup_l7.argtypes = [p, p, p, u64, u64, u32, u64, ct.c_char_p,
                                 u32, p, p, u64]
    lib.df_rollup_insert.restype = ct.c_int
    lib.df_rollup_insert.argtypes = [p, p, p, u32, u32, u32, p, p, p, u32,
                                     p, u64]
    lib.df_gather_records.restype = ct.c_int
    lib.df_gather_records.argtypes = [p, p, p, p, p, u32, p, u64]
    lib.df_kg_build.restype = ct.c_int
    lib.df_kg_build.argtypes = [p, p, u32, p, p, u32, u64]
    lib.df_intern_many.restype = ct.c_int
    lib.df_intern_many.argtypes = [p, p, p, p, u32, u32, u64, u64, p, u32,
                                   p, p, u32, p, u64, u64, u64]
    lib.df_intern_attrs.restype = ct.c_int
    lib.df_intern_attrs.argtypes = [p, p, p, u32, u64, u64, u64, p, u32,
                                    p, p, u32, p, p, u64]
    lib.df_pool_lens.restype = ct.c_int
    lib.df_pool_lens.argtypes = [p, p, u32, u32, u64, u64, p, u64]
    lib.df_pool_gather.restype = ct.c_int
    lib.df_pool_gather.argtypes = [p, p, p, u32, u32, u64, u64, p, p, u64, p, p, u64, u64, u64]
    lib.df_query_agg.restype = ct.c_int
    lib.df_query_agg.argtypes = [p, p, p, p, p, p, u32, p, p, p, p, p, p,
                                 u64, u64, p, u32, u64, p, p, p, u32, u64]
    lib.df_qpart_agg.restype = ct.c_int
    lib.df_qpart_agg.argtypes = [p, p, p, p, p, p, u32, p, p, p, p, p, p,
                                 u64, u64, p, u32, u64, p, p, p, p, p, p,
                                 u32, u64]
    lib.df_query_select.restype = ct.c_int
    lib.df_query_select.argtypes = [p, p, p, p, p, p, u32, p, p, p, p, p,
                                    p, u64, u64, p, u32, u64, p, p, u32,
                                    u64]
    lib.df_spec_sizes.restype = ct.c_int
    lib.df_spec_sizes.argtypes = [p, p, p, p]
    lib.df_sort_u64.restype = ct.c_int
    lib.df_sort_u64.argtypes = [p, p, u32, p, ct.POINTER(ct.c_uint64), u64]
    lib.df_pack_bits.restype = ct.c_int
    lib.df_pack_bits.argtypes = [p, u32, u32, u32, p, u32, u64]
    lib.df_unpack_bits.restype = ct.c_int
    lib.df_unpack_bits.argtypes = [p, u32, u32, u32, p, u64]


def gpu() -> ct.CDLL:
    """Load the HIP kernel library; raises if missing (no silent fallback)."""
    global _gpu_lib
    if _gpu_lib is None:
        path = OPS_DIR / "libdfgpu.so"
        if not path.exists():
            raise RuntimeError(
                "libdfgpu.so not built — run deepflow_amd/ops/build.py "
                "(GPU ops never fall back to eager)")
        lib = ct.CDLL(str(path))
        _decl_gpu(lib)
        _gpu_lib = lib
    return _gpu_lib


def check(rc: int, what: str) -> None:
    if rc != 0:
        raise RuntimeError(f"HIP error {rc} in {what}")
