"""Golden tests: the C++ fast generator must be byte-identical to the Python
reference generator, and the offset scanner must agree with framing.py."""
import ctypes as ct

import numpy as np
import pytest

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.wire import framing
from deepflow_amd.ops import native


@pytest.fixture(scope="module")
def cpu_lib():
    return native.cpu()


def _gen_native(cpu_lib, cfg, i0, n):
    c = native.span_cfg_c(cfg)
    need = cpu_lib.df_gen_spans(ct.byref(c), i0, n, None, 0)
    buf = np.zeros(need, dtype=np.uint8)
    wrote = cpu_lib.df_gen_spans(ct.byref(c), i0, n,
                                 buf.ctypes.data_as(ct.c_void_p), need)
    assert wrote == need
    return buf.tobytes()


def test_span_payload_golden(cpu_lib):
    cfg = SpanGenConfig(n=64, seed=7, tag_cardinality=1000, n_attrs=4)
    py = gen_span_payload(cfg)
    cc = _gen_native(cpu_lib, cfg, 0, 64)
    assert py == cc


def test_span_payload_golden_offsets(cpu_lib):
    cfg = SpanGenConfig(n=32, seed=11, tag_cardinality=50, n_attrs=2,
                        err_rate_pct=50)
    py = gen_span_payload(cfg)
    cc = _gen_native(cpu_lib, cfg, 0, 32)
    assert py == cc
    # indexed variant agrees with the scanner
    c = native.span_cfg_c(cfg)
    need = cpu_lib.df_gen_spans(ct.byref(c), 0, 32, None, 0)
    buf = np.zeros(need, dtype=np.uint8)
    offs = np.zeros(32, dtype=np.uint32)
    lens = np.zeros(32, dtype=np.uint32)
    cpu_lib.df_gen_spans_indexed(ct.byref(c), 0, 32,
                                 buf.ctypes.data_as(ct.c_void_p), need,
                                 offs.ctypes.data_as(ct.c_void_p),
                                 lens.ctypes.data_as(ct.c_void_p))
    expected = framing.scan_record_offsets(buf.tobytes())
    assert [(int(o), int(l)) for o, l in zip(offs, lens)] == expected


def test_scan_offsets_native(cpu_lib):
    cfg = SpanGenConfig(n=16, seed=3)
    payload = gen_span_payload(cfg)
    buf = np.frombuffer(payload, dtype=np.uint8)
    offs = np.zeros(16, dtype=np.uint32)
    lens = np.zeros(16, dtype=np.uint32)
    n = cpu_lib.df_scan_offsets(buf.ctypes.data_as(ct.c_void_p), len(payload),
                                offs.ctypes.data_as(ct.c_void_p),
                                lens.ctypes.data_as(ct.c_void_p), 16)
    assert n == 16
    expected = framing.scan_record_offsets(payload)
    assert [(int(o), int(l)) for o, l in zip(offs, lens)] == expected


def test_gen_chunked_matches_full(cpu_lib):
    cfg = SpanGenConfig(n=40, seed=5)
    full = _gen_native(cpu_lib, cfg, 0, 40)
    a = _gen_native(cpu_lib, cfg, 0, 17)
    b = _gen_native(cpu_lib, cfg, 17, 23)
    assert a + b == full


def test_parallel_generator_identical(cpu_lib):
    import ctypes as ct
    cfg = SpanGenConfig(n=500, seed=21, tag_cardinality=100)
    c = native.span_cfg_c(cfg)
    need = cpu_lib.df_gen_spans_parallel(ct.byref(c), 0, 500, None, 0, None, None)
    buf = np.zeros(int(need), dtype=np.uint8)
    offs = np.zeros(500, dtype=np.uint32)
    lens = np.zeros(500, dtype=np.uint32)
    cpu_lib.df_gen_spans_parallel(ct.byref(c), 0, 500,
                                  buf.ctypes.data_as(ct.c_void_p), need,
                                  offs.ctypes.data_as(ct.c_void_p),
                                  lens.ctypes.data_as(ct.c_void_p))
    assert buf.tobytes() == gen_span_payload(cfg)
    expected = framing.scan_record_offsets(buf.tobytes())
    assert [(int(o), int(l)) for o, l in zip(offs, lens)] == expected


def test_span_payload_golden_ipv6(cpu_lib):
    cfg = SpanGenConfig(n=48, seed=3, tag_cardinality=100, n_attrs=2,
                        ip6_rate_pct=40)
    py = gen_span_payload(cfg)
    cc = _gen_native(cpu_lib, cfg, 0, 48)
    assert py == cc
