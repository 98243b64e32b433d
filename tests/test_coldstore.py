"""Cold-segment bit-pack codec: roundtrip + query-over-cold equivalence
(CPU reference; the GPU kernel twins are in test_coldstore_gpu.py)."""
import pytest
import torch

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.ingest import L7IngestPipeline
from deepflow_amd.query.engine import QueryEngine
from deepflow_amd.store import coldstore as C

CFG = SpanGenConfig(n=2000, seed=33, tag_cardinality=50, n_attrs=2,
                    n_ips=64, n_services=4, n_resources=12)


def test_pack_roundtrip_cpu():
    for bits in (1, 3, 7, 12, 17, 24, 31, 32):
        hi = (1 << bits) - 1
        vals = torch.randint(0, hi + 1 if hi < 2**31 else 2**31,
                             (997,), dtype=torch.int64)
        if bits == 32:
            vals = (vals * 2677) & 0xFFFFFFFF
        packed = C.pack_stream(vals.to(torch.int32), 0, bits)
        assert packed.numel() == (997 * bits + 31) // 32
        out = C.unpack_stream(packed, 997, 0, bits)
        got = out.to(torch.int64) & 0xFFFFFFFF
        assert torch.equal(got, vals & 0xFFFFFFFF), bits


def _pipe():
    p = L7IngestPipeline(device="cpu", segment_rows=1 << 11,
                         dict_capacity=1 << 12,
                         time_base_s=CFG.base_time_ns // 10**9)
    p.ingest_frame_payload(gen_span_payload(CFG))       # fills segment 0
    p.ingest_frame_payload(gen_span_payload(CFG))       # rolls to segment 1
    return p


def test_segment_compress_materialize_identical():
    pipe = _pipe()
    seg = pipe.segments.segments[0]
    n = seg.n_rows
    snap = {k: getattr(seg, k)[..., :n].clone()
            for k in ("u64", "u32", "u8", "did", "str_lens",
                      "attr_start", "attr_cnt")}
    rowref = seg.str_rowref[:n].clone()
    pool = seg.pool[: seg.pool_len].clone()
    attr_pool = seg.attr_pool[: seg.attr_pool_len].clone()

    cseg = C.CompressedL7Segment(seg)
    ratio = (seg.stored_bytes_per_row() * n) / cseg.compressed_bytes()
    assert ratio > 1.5, ratio  # bit-packing must actually pay

    from deepflow_amd.store.segment import L7Segment, SegmentSet
    scratch = L7Segment(seg.capacity, "cpu")
    cseg.materialize(scratch)
    for k, want in snap.items():
        got = getattr(scratch, k)[..., :n]
        assert torch.equal(got, want), k
    assert torch.equal(scratch.str_rowref[:n], rowref)
    assert torch.equal(scratch.pool[: scratch.pool_len], pool)
    assert torch.equal(scratch.attr_pool[: scratch.attr_pool_len], attr_pool)


QUERIES = [
    "SELECT Count(*) AS c FROM l7_flow_log",
    "SELECT l7_protocol, Count(*) AS c, Avg(response_duration) AS a "
    "FROM l7_flow_log GROUP BY l7_protocol ORDER BY c DESC",
    "SELECT request_resource, Count(*) AS c FROM l7_flow_log "
    "WHERE response_status = 'Server Error' GROUP BY request_resource "
    "ORDER BY c DESC LIMIT 5",
]


def test_query_over_cold_equals_hot():
    pipe = _pipe()
    eng = QueryEngine(pipe, device="cpu")
    want = [eng.query(q) for q in QUERIES]
    demoted = pipe.segments.demote_oldest()
    assert demoted
    assert pipe.segments.cold and len(pipe.segments.segments) >= 1
    got = [eng.query(q) for q in QUERIES]
    for w, g in zip(want, got):
        assert w == g
    # string hydration still works on materialized cold rows
    r = eng.query("SELECT request_resource FROM l7_flow_log LIMIT 3")
    assert all(v[0].startswith("/") for v in r["values"])


def test_demote_recycles_buffers():
    pipe = _pipe()
    segs_before = len(pipe.segments.segments)
    free_before = len(pipe.segments._free)
    pipe.segments.demote_oldest()
    assert len(pipe.segments.segments) == segs_before - 1
    assert len(pipe.segments._free) == free_before + 1
    assert pipe.segments.n_rows == 2 * CFG.n  # rows still accounted


def test_two_tier_watermark():
    """Over hot_max_bytes -> compress oldest; over max_bytes -> drop
    oldest cold."""
    pipe = _pipe()                      # 2 segments of 2000 rows
    segset = pipe.segments
    seg_bytes = segset.seg_alloc_bytes(segset.segments[0])
    segset.hot_max_bytes = int(seg_bytes * 1.5)   # hot window: 1 segment
    from deepflow_amd.gen.spans import gen_span_payload
    pipe.ingest_frame_payload(gen_span_payload(CFG))  # rolls a 3rd segment
    assert len(segset.cold) >= 1        # oldest got compressed, not lost
    assert segset.n_rows == 3 * CFG.n
    r_cold = segset.cold[0].compressed_bytes()
    # packed form is well under the allocated segment footprint (the raw
    # string pool survives uncompressed, so the bound is conservative)
    assert r_cold < seg_bytes * 0.7


def test_lazy_cold_materialization():
    """Back-to-back queries with different column sets over cold data
    (lazy per-column decompression must not leak stale columns)."""
    pipe = _pipe()
    eng = QueryEngine(pipe, device="cpu")
    want = [
        eng.query("SELECT l7_protocol, Count(*) AS c FROM l7_flow_log "
                  "GROUP BY l7_protocol ORDER BY c DESC"),
        eng.query("SELECT request_domain, Sum(response_duration) AS s "
                  "FROM l7_flow_log GROUP BY request_domain "
                  "ORDER BY s DESC LIMIT 5"),
        eng.query("SELECT request_resource, trace_id FROM l7_flow_log "
                  "LIMIT 5"),
    ]
    pipe.segments.demote_oldest()
    got = [
        eng.query("SELECT l7_protocol, Count(*) AS c FROM l7_flow_log "
                  "GROUP BY l7_protocol ORDER BY c DESC"),
        eng.query("SELECT request_domain, Sum(response_duration) AS s "
                  "FROM l7_flow_log GROUP BY request_domain "
                  "ORDER BY s DESC LIMIT 5"),
        eng.query("SELECT request_resource, trace_id FROM l7_flow_log "
                  "LIMIT 5"),
    ]
    assert want == got
    # explicit needed-set restriction: only one u8 column decompressed
    from deepflow_amd.query import spec as Q
    segs = pipe.segments.scan_list(
        needed={Q.SRC_U8: {0}})
    assert segs[0].n_rows == CFG.n


def test_l4_cold_tier():
    """L4 segments demote/query through the same codec (no dict/attr
    blocks)."""
    from deepflow_amd.agent import Agent
    from deepflow_amd.agent.packets import http_session
    from deepflow_amd.ingest.l4_pipeline import L4IngestPipeline
    from deepflow_amd.query.engine import QueryEngine
    from deepflow_amd.ingest import L7IngestPipeline
    import numpy as np
    l4 = L4IngestPipeline(device="cpu", segment_rows=1 << 6, time_base_s=0)
    a = Agent(vtap_id=1)
    for i in range(150):
        for frame, ts in http_session(0x0A000001 + i, 0x0A000002,
                                      sport=40000 + i, t0=10**9):
            a.packet(frame, ts)
    a.tick(1 << 62)
    payload = a.drain(0)
    from deepflow_amd.ops import native
    import ctypes as ct
    lib = native.cpu()
    buf = np.frombuffer(payload, dtype=np.uint8)
    offs = np.zeros(4096, dtype=np.uint32)
    lens = np.zeros(4096, dtype=np.uint32)
    n = int(lib.df_scan_offsets(buf.ctypes.data_as(ct.c_void_p), len(buf),
                                offs.ctypes.data_as(ct.c_void_p),
                                lens.ctypes.data_as(ct.c_void_p), 4096))
    # chunk so several segments fill (segment_rows = 64)
    for i in range(0, n, 60):
        l4.ingest(buf, offs[i:i + 60].copy(), lens[i:i + 60].copy())
    assert len(l4.segments.segments) >= 2
    l7 = L7IngestPipeline(device="cpu", segment_rows=1 << 8,
                          dict_capacity=1 << 10, time_base_s=0)
    eng = QueryEngine(l7, device="cpu", l4_pipeline=l4)
    q = "SELECT Count(*) AS c, Sum(byte_tx) AS tx FROM l4_flow_log"
    want = eng.query(q)
    assert l4.segments.demote_oldest()
    got = eng.query(q)
    assert want == got
