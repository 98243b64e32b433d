"""Checkpoint/restore of the hot store: queries after restore match the
original; the layout-migration hook upgrades old manifests (ckissu role)."""
import pytest

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.ingest import L7IngestPipeline
from deepflow_amd.query.engine import QueryEngine
from deepflow_amd.store import checkpoint as CK

CFG = SpanGenConfig(n=1200, seed=21, tag_cardinality=40, n_attrs=2,
                    n_ips=64, n_services=4, n_resources=10)


def _pipe():
    p = L7IngestPipeline(device="cpu", segment_rows=1 << 11,
                         dict_capacity=1 << 12,
                         time_base_s=CFG.base_time_ns // 10**9)
    p.ingest_frame_payload(gen_span_payload(CFG))
    return p


QUERIES = [
    "SELECT Count(*) AS c FROM l7_flow_log",
    "SELECT request_resource, Count(*) AS c, Avg(response_duration) AS a "
    "FROM l7_flow_log GROUP BY request_resource ORDER BY c DESC LIMIT 5",
    "SELECT l7_protocol, Count(*) AS c FROM l7_flow_log "
    "WHERE request_domain = 'svc-001.example.com' GROUP BY l7_protocol",
]


def test_checkpoint_roundtrip(tmp_path):
    pipe = _pipe()
    eng = QueryEngine(pipe, device="cpu")
    want = [eng.query(q) for q in QUERIES]
    path = str(tmp_path / "shard0.ckpt")
    CK.save_l7(pipe, path)

    fresh = L7IngestPipeline(device="cpu", segment_rows=1 << 11,
                             dict_capacity=1 << 12,
                             time_base_s=CFG.base_time_ns // 10**9)
    n = CK.load_l7(fresh, path)
    assert n == CFG.n
    eng2 = QueryEngine(fresh, device="cpu")
    got = [eng2.query(q) for q in QUERIES]
    assert want == got
    # ingest continues after restore (dictionary state intact)
    fresh.ingest_frame_payload(gen_span_payload(CFG))
    r = eng2.query("SELECT Count(*) AS c FROM l7_flow_log")
    assert r["values"] == [[2 * CFG.n]]
    # rollups restored too
    assert fresh.metrics.rows()


def test_migration_hook(tmp_path):
    import torch
    pipe = _pipe()
    path = str(tmp_path / "old.ckpt")
    CK.save_l7(pipe, path)
    payload = torch.load(path, weights_only=False)
    payload["layout_version"] -= 1          # pretend it is one version old
    torch.save(payload, path)

    calls = []

    from deepflow_amd.store import l7_schema as S

    @CK.register_migration(S.LAYOUT_VERSION - 1)
    def up(p):
        calls.append(1)
        p["layout_version"] = S.LAYOUT_VERSION
        return p

    try:
        fresh = L7IngestPipeline(device="cpu", segment_rows=1 << 11,
                                 dict_capacity=1 << 12,
                                 time_base_s=CFG.base_time_ns // 10**9)
        assert CK.load_l7(fresh, path) == CFG.n
        assert calls == [1]
    finally:
        CK.MIGRATIONS.pop(S.LAYOUT_VERSION - 1, None)

    # unknown version refuses loudly
    payload["layout_version"] = 0
    torch.save(payload, path)
    fresh2 = L7IngestPipeline(device="cpu", segment_rows=1 << 11,
                              dict_capacity=1 << 12)
    with pytest.raises(RuntimeError, match="no migration"):
        CK.load_l7(fresh2, path)


def test_checkpoint_with_cold_tier(tmp_path):
    """Demoted (bit-packed) segments survive checkpoint/restore packed."""
    pipe = L7IngestPipeline(device="cpu", segment_rows=1 << 10,
                            dict_capacity=1 << 12,
                            time_base_s=CFG.base_time_ns // 10**9)
    small = SpanGenConfig(n=900, seed=21, tag_cardinality=40, n_attrs=2,
                          n_ips=64, n_services=4, n_resources=10)
    pipe.ingest_frame_payload(gen_span_payload(small))
    pipe.ingest_frame_payload(gen_span_payload(small))
    assert pipe.segments.demote_oldest()
    eng = QueryEngine(pipe, device="cpu")
    q = ("SELECT request_resource, Count(*) AS c, "
         "Avg(response_duration) AS a FROM l7_flow_log "
         "GROUP BY request_resource ORDER BY c DESC")
    want = eng.query(q)
    path = str(tmp_path / "cold.ckpt")
    CK.save_l7(pipe, path)
    fresh = L7IngestPipeline(device="cpu", segment_rows=1 << 10,
                             dict_capacity=1 << 12,
                             time_base_s=CFG.base_time_ns // 10**9)
    assert CK.load_l7(fresh, path) == 1800
    assert len(fresh.segments.cold) == 1           # restored still packed
    assert fresh.segments.cold[0].compressed_bytes() > 0
    got = QueryEngine(fresh, device="cpu").query(q)
    assert want == got


def test_load_validates_time_base_and_capacity(tmp_path):
    pipe = _pipe()
    path = str(tmp_path / "s.ckpt")
    CK.save_l7(pipe, path)
    # a pipeline built with a different time base gets the saved one back
    fresh = L7IngestPipeline(device="cpu", segment_rows=1 << 11,
                             dict_capacity=1 << 12, time_base_s=0)
    CK.load_l7(fresh, path)
    assert fresh.time_base_s == pipe.time_base_s
    # mismatched dict capacity is a hard error (slot ids are positional)
    bad = L7IngestPipeline(device="cpu", segment_rows=1 << 11,
                           dict_capacity=1 << 10,
                           time_base_s=pipe.time_base_s)
    with pytest.raises(RuntimeError, match="dictionary capacity"):
        CK.load_l7(bad, path)
    # smaller segment_rows is a hard error too
    small = L7IngestPipeline(device="cpu", segment_rows=1 << 9,
                             dict_capacity=1 << 12,
                             time_base_s=pipe.time_base_s)
    with pytest.raises(RuntimeError, match="segment_rows"):
        CK.load_l7(small, path)


def test_server_checkpoint_wiring(tmp_path):
    """save_checkpoint/load_checkpoint on the server persist the hot store
    AND the controller registry (agents keep ids/GPIDs across restart)."""
    from deepflow_amd.server import DeepflowServer
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 11,
                         dict_capacity=1 << 12,
                         time_base_s=CFG.base_time_ns // 10**9)
    srv.l7.ingest_frame_payload(gen_span_payload(CFG))
    srv.controller.sync(7, hostname="host-a", ip="10.0.0.7")
    g = srv.controller.genesis_report(
        7, [{"pid": 123, "name": "nginx"}], [])
    gpid = g["gpids"][123]
    srv.controller.set_group_config("default", {"sync_interval": 30})
    want = srv.engine.query(QUERIES[1])
    out = srv.save_checkpoint(str(tmp_path))
    assert out["rows"] == CFG.n and out["agents"] == 1

    srv2 = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 11,
                          dict_capacity=1 << 12,
                          time_base_s=CFG.base_time_ns // 10**9)
    restored = srv2.load_checkpoint(str(tmp_path))
    assert restored["l7_rows"] == CFG.n and restored["controller"]
    assert srv2.engine.query(QUERIES[1]) == want
    # GPID allocation is stable across restart (metadb role)
    assert srv2.controller.lookup_gpid(7, 123) == gpid
    g2 = srv2.controller.genesis_report(
        7, [{"pid": 123, "name": "nginx"}, {"pid": 456, "name": "redis"}], [])
    assert g2["gpids"][123] == gpid
    assert g2["gpids"][456] != gpid
    assert srv2.controller.group_configs["default"]["sync_interval"] == 30


def test_scratch_pool_released_after_cold_query():
    """Cold-tier query scratch returns to the free-list (bounded), and
    scratch is counted by the watermark accounting."""
    pipe = L7IngestPipeline(device="cpu", segment_rows=1 << 8,
                            dict_capacity=1 << 12,
                            time_base_s=CFG.base_time_ns // 10**9)
    import dataclasses
    small = dataclasses.replace(CFG, n=200)
    for i in range(6):
        pipe.ingest_frame_payload(gen_span_payload(
            dataclasses.replace(small, seed=21 + i)))
    segs = pipe.segments
    # force all but the tail segment into the cold tier
    while segs.demote_oldest():
        pass
    assert getattr(segs, "cold", [])
    eng = QueryEngine(pipe, device="cpu")
    r = eng.query("SELECT Count(*) AS c FROM l7_flow_log")
    assert r["values"][0][0] == 6 * 200
    # after the query the scratch pool is empty and its segments are free
    assert not getattr(segs, "_scratch", [])
    assert segs._free


def test_l4_and_prom_checkpoint_roundtrip(tmp_path):
    """Server checkpoints now cover the L4 flow store and the prometheus
    sample columns (reference durable state = ClickHouse tables)."""
    from deepflow_amd.server import DeepflowServer
    from deepflow_amd.gen.flows import FlowGenConfig, gen_flow_payload
    from deepflow_amd.wire import framing

    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12, time_base_s=0)
    cfg = FlowGenConfig(n=40, seed=6)
    srv._on_l4(framing.FrameHeader(msg_type=framing.MSG_TAGGEDFLOW),
               __import__("numpy").frombuffer(gen_flow_payload(cfg),
                                              dtype="uint8"))
    srv.prom.ingest_labeled_samples([
        ("up", {"job": "api"}, 1000, 1.0),
        ("up", {"job": "db"}, 2000, 0.0)])
    d = str(tmp_path / "ck")
    srv.save_checkpoint(d)

    srv2 = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                          dict_capacity=1 << 12, time_base_s=0)
    out = srv2.load_checkpoint(d)
    assert out["l4_rows"] == 40
    assert out["prom"]
    r = srv2.engine.query("SELECT COUNT(1) FROM l4_flow_log")
    assert r["values"][0][0] == 40
    series = srv2.prom.series_for("up", [("job", "=", "api")])
    assert series and series[0]["samples"] == {1: 1.0}


@pytest.mark.gpu
def test_l4_checkpoint_roundtrip_gpu(tmp_path):
    from deepflow_amd.ingest.l4_pipeline import L4IngestPipeline
    from deepflow_amd.store import checkpoint as ck
    from deepflow_amd.gen.flows import FlowGenConfig, gen_flow_payload
    import numpy as np

    cfg = FlowGenConfig(n=500, seed=9)
    pipe = L4IngestPipeline(device="cuda", segment_rows=1 << 12,
                            time_base_s=0)
    pipe.ingest_frame_payload(gen_flow_payload(cfg))
    path = str(tmp_path / "l4.ckpt")
    ck.save_l4(pipe, path)
    pipe2 = L4IngestPipeline(device="cuda", segment_rows=1 << 12,
                             time_base_s=0)
    assert ck.load_l4(pipe2, path) == 500
    a = pipe.segments.segments[0]
    b = pipe2.segments.segments[0]
    import torch
    assert torch.equal(a.u64[:, :500], b.u64[:, :500])
    assert torch.equal(a.u32[:, :500], b.u32[:, :500])
    assert a.pool_len == b.pool_len
