"""flow_metrics table family: exact keys (no masked-pack collisions),
_map variants, traffic_policy, datasource intervals, doc insert merge.
Reference: server/libs/flow-metrics/tag.go:443-523."""
import dataclasses

import pytest

from deepflow_amd.gen.flows import FlowGenConfig, gen_flow_dict, \
    gen_flow_payload
from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.ingest.l4_pipeline import L4IngestPipeline
from deepflow_amd.ingest import L7IngestPipeline
from deepflow_amd.query.engine import QueryEngine
from deepflow_amd.store.metrics import RollupTable, TableDef

BASE = 1_700_000_000

CFG = FlowGenConfig(n=600, seed=5, n_ips=64, n_epcs=8, acl_rate_pct=25,
                    n_acls=3, ip6_rate_pct=20)


@pytest.fixture(scope="module")
def l4pipe():
    p = L4IngestPipeline(device="cpu", segment_rows=1 << 10,
                         time_base_s=BASE)
    p.ingest_frame_payload(gen_flow_payload(CFG))
    return p


def test_network_tables_exact(l4pipe):
    """network.1s groups match a brute-force rollup of the generated
    flows (exact keys, all tag columns)."""
    want = {}
    for i in range(CFG.n):
        f = gen_flow_dict(CFG, i)["flow"]
        t = f["start_time"] // 10**9 - BASE
        key = (t, f["flow_key"]["vtap_id"],
               f["metrics_peer_src"]["l3_epc_id"],
               f["flow_key"]["proto"], f["flow_key"]["port_dst"],
               f["flow_key"]["tap_type"])
        acc = want.setdefault(key, [0, 0])
        acc[0] += f["metrics_peer_src"]["byte_count"]
        acc[1] += f["metrics_peer_dst"]["byte_count"]
    rows = l4pipe.rollups.get("network.1s").rows()
    got = {(r["time"] - BASE, r["vtap_id"], r["l3_epc_id"], r["protocol"],
            r["server_port"], r["tap_type"]): [r["byte_tx"], r["byte_rx"]]
           for r in rows}
    assert got == want


def test_network_map_has_both_endpoints(l4pipe):
    rows = l4pipe.rollups.get("network_map.1s").rows()
    assert rows
    r = rows[0]
    for k in ("ip_0", "ip_1", "l3_epc_id_0", "l3_epc_id_1", "protocol",
              "server_port"):
        assert k in r
    # v4 rows carry dotted addresses; v6 flows have zeroed v4 fields
    assert any(r["ip_0"].startswith("10.") for r in rows)
    assert any(r["ip_0"] == "0.0.0.0" for r in rows)  # the v6 flows


def test_traffic_policy_only_acl_rows(l4pipe):
    n_acl = sum(1 for i in range(CFG.n)
                if gen_flow_dict(CFG, i)["flow"].get("acl_gids"))
    assert n_acl > 0
    rows = l4pipe.rollups.get("traffic_policy.1m").rows()
    assert rows
    assert all(r["acl_gid"] != 0 for r in rows)
    gids = {r["acl_gid"] for r in rows}
    assert gids == {1, 2, 3}
    # every ACL-matched flow is accounted: sum of new_flow == n_acl? flows
    # all have close_type=1 -> closed_flow counts flows exactly
    assert sum(r["closed_flow"] for r in rows) == n_acl


def test_1m_buckets_are_minute_aligned(l4pipe):
    rows = l4pipe.rollups.get("network.1m").rows()
    assert rows
    assert all((r["time"] - BASE) % 60 == 0 for r in rows)
    s1 = sum(r["byte_tx"] for r in l4pipe.rollups.get("network.1s").rows())
    assert sum(r["byte_tx"] for r in rows) == s1


def test_no_epc_collision_regression():
    """Round-1 defect: epc ids differing only above bit 16 merged
    (k_agg_net1s packed `epc & 0xFFFF`). Exact keys must keep them apart."""
    p = L4IngestPipeline(device="cpu", segment_rows=1 << 10,
                         time_base_s=BASE)
    payload = gen_flow_payload(dataclasses.replace(CFG, n=4, acl_rate_pct=0,
                                                   ip6_rate_pct=0))
    p.ingest_frame_payload(payload)
    seg = p.segments.segments[0]
    import torch
    epc_idx = 3  # l3_epc_id_0
    # rewrite two rows' epc to values that collide under a 16-bit mask
    seg.u32[epc_idx, 0] = 5
    seg.u32[epc_idx, 1] = 5 + (1 << 16)
    t = p.rollups.get("network.1s")
    t.table.clear()
    t.update(seg, 0, 2)
    epcs = sorted(r["l3_epc_id"] for r in t.rows())
    assert epcs == [5, 5 + (1 << 16)]


def test_l4_ipv6_queryable(l4pipe):
    eng = QueryEngine(L7IngestPipeline(device="cpu", segment_rows=1 << 8,
                                       dict_capacity=1 << 10,
                                       time_base_s=BASE),
                      device="cpu", l4_pipeline=l4pipe)
    n_v6 = sum(1 for i in range(CFG.n)
               if "ip6_src" in gen_flow_dict(CFG, i)["flow"]["flow_key"])
    assert n_v6 > 0
    r = eng.query("SELECT ip6_0, ip6_1 FROM l4_flow_log "
                  "WHERE ip6_0 != '::' LIMIT 5")
    vals = [v for v in r["values"] if v[0]]
    assert vals and vals[0][0].startswith("2001:db8::")
    one = vals[0][0]
    r2 = eng.query("SELECT Count(*) AS c FROM l4_flow_log "
                   f"WHERE ip6_0 = '{one}'")
    assert r2["values"][0][0] >= 1


def test_application_tables_and_engine_routing():
    scfg = SpanGenConfig(n=500, seed=9, tag_cardinality=50, n_attrs=1,
                         n_ips=32, n_services=4, n_resources=8)
    p = L7IngestPipeline(device="cpu", segment_rows=1 << 10,
                         dict_capacity=1 << 12,
                         time_base_s=scfg.base_time_ns // 10**9)
    p.ingest_frame_payload(gen_span_payload(scfg))
    eng = QueryEngine(p, device="cpu")
    r = eng.query("SELECT Sum(request) AS r FROM application.1s")
    total = r["values"][0][0]
    assert total == scfg.n
    # 1m table agrees in total
    r = eng.query("SELECT Sum(request) AS r FROM application.1m")
    assert r["values"][0][0] == total
    # map table carries the ip pair
    r = eng.query("SELECT ip_1, Sum(request) AS r FROM application_map.1s "
                  "GROUP BY ip_1 ORDER BY r DESC LIMIT 3")
    assert r["values"] and r["values"][0][0].startswith("10.")
    # derived datasource (1h) rides the 1m table
    r = eng.query("SELECT Sum(request) AS r FROM application.1h")
    assert r["values"][0][0] == total


def test_doc_insert_merges_exactly():
    td = TableDef("t.agent", "doc", 1, ("vtap_id", "x"), "app", 0)
    t = RollupTable(td, 0, device="cpu")
    t.insert([(5, 1, 7), (5, 1, 7), (5, 2, 7)],
             [[1, 0, 0, 0, 10, 1, 10],
              [2, 1, 0, 0, 30, 1, 30],
              [4, 0, 1, 0, 0, 0, 0]])
    rows = t.rows()
    assert len(rows) == 2
    merged = [r for r in rows if r["vtap_id"] == 1][0]
    assert merged["request"] == 3 and merged["response"] == 1
    assert merged["rrt_sum"] == 40 and merged["rrt_max"] == 30  # max op


def test_rollup_interval_flush_merges_exactly():
    """Archive-and-reset flush: a group updated before AND after a flush
    reads back as one exactly-merged row, identical to a never-flushed
    table."""
    from deepflow_amd.gen.spans import SpanGenConfig, gen_span_payload
    from deepflow_amd.ingest import L7IngestPipeline

    cfg = SpanGenConfig(n=4000, seed=21, tag_cardinality=500, n_ips=64,
                        n_services=8)
    payload = gen_span_payload(cfg)
    a = L7IngestPipeline(device="cpu", segment_rows=1 << 13,
                         time_base_s=cfg.base_time_ns // 10**9)
    b = L7IngestPipeline(device="cpu", segment_rows=1 << 13,
                         time_base_s=cfg.base_time_ns // 10**9)
    a.ingest_frame_payload(payload)
    a.rollups.flush()           # everything moves to the archive
    a.ingest_frame_payload(payload)
    a.rollups.flush()
    a.ingest_frame_payload(payload)
    for _ in range(3):
        b.ingest_frame_payload(payload)
    for name in ("application.1s", "application.1m",
                 "application_map.1s"):
        ra = a.rollups.get(name).rows()
        rb = b.rollups.get(name).rows()
        assert ra == rb, name


def test_flow_metrics_sql_after_flush():
    """DF-SQL over the flow_metrics family still sees every group after
    interval flushes (archive + live merge feeds the engine)."""
    from deepflow_amd.gen.spans import SpanGenConfig, gen_span_payload
    from deepflow_amd.ingest import L7IngestPipeline
    from deepflow_amd.query import QueryEngine

    cfg = SpanGenConfig(n=3000, seed=31, tag_cardinality=200, n_ips=32)
    pipe = L7IngestPipeline(device="cpu", segment_rows=1 << 13,
                            time_base_s=cfg.base_time_ns // 10**9)
    payload = gen_span_payload(cfg)
    pipe.ingest_frame_payload(payload)
    pipe.rollups.flush()
    pipe.ingest_frame_payload(payload)
    eng = QueryEngine(pipe, device="cpu")
    r = eng.query("SELECT Sum(request) AS s FROM application.1s")
    assert r["values"][0][0] == 6000
    r = eng.query("SELECT Sum(request) AS s FROM application.1m")
    assert r["values"][0][0] == 6000
