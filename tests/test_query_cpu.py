"""DF-SQL engine tests on the CPU pipeline (oracle checked against direct
pb-truth recomputation)."""
import pytest

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_dict, gen_span_payload
from deepflow_amd.ingest import L7IngestPipeline
from deepflow_amd.query import QueryEngine
from deepflow_amd.store.kg import KnowledgeGraphTable, default_platform

N = 300
CFG = SpanGenConfig(n=N, seed=101, tag_cardinality=40, n_attrs=2,
                    n_ips=64, n_services=4, n_resources=10)


@pytest.fixture(scope="module")
def eng():
    kg = KnowledgeGraphTable(capacity_pow2=1 << 12, device="cpu")
    kg.update(default_platform(CFG))
    p = L7IngestPipeline(device="cpu", segment_rows=1 << 10, kg=kg,
                        dict_capacity=1 << 12,
                        time_base_s=CFG.base_time_ns // 10**9)
    p.ingest_frame_payload(gen_span_payload(CFG))
    return QueryEngine(p, device="cpu")


def truth():
    return [gen_span_dict(CFG, i) for i in range(N)]


def test_count_all(eng):
    r = eng.query("SELECT Count(*) AS cnt FROM l7_flow_log")
    assert r["values"] == [[N]]


def test_group_by_domain(eng):
    r = eng.query(
        "SELECT request_domain, Count(*) AS cnt FROM l7_flow_log "
        "GROUP BY request_domain ORDER BY cnt DESC")
    got = {row[0]: row[1] for row in r["values"]}
    want = {}
    for t in truth():
        want[t["req"]["domain"]] = want.get(t["req"]["domain"], 0) + 1
    assert got == want


def test_where_string_filter(eng):
    t0 = truth()
    dom = t0[0]["req"]["domain"]
    r = eng.query(
        f"SELECT Count(*) AS cnt FROM l7_flow_log WHERE request_domain = '{dom}'")
    want = sum(1 for t in t0 if t["req"]["domain"] == dom)
    assert r["values"] == [[want]]


def test_where_unknown_string(eng):
    r = eng.query(
        "SELECT Count(*) AS c FROM l7_flow_log WHERE request_domain = 'nope'")
    assert r["values"] == []


def test_avg_and_max_duration(eng):
    r = eng.query(
        "SELECT Avg(response_duration) AS a, Max(response_duration) AS m "
        "FROM l7_flow_log")
    rrts = [t["base"]["head"]["rrt"] for t in truth()]
    a, m = r["values"][0]
    assert m == max(rrts)
    assert abs(a - sum(rrts) / len(rrts)) < 1e-6


def test_status_filter_numeric_and_group(eng):
    r = eng.query(
        "SELECT response_status, Count(*) AS c FROM l7_flow_log "
        "GROUP BY response_status")
    got = {row[0]: row[1] for row in r["values"]}
    errs = sum(1 for t in truth() if t["resp"]["status"] == 3)
    assert got.get("Server Error", 0) == errs
    assert got.get("Success", 0) == N - errs


def test_time_bucket_group(eng):
    r = eng.query(
        "SELECT time(60), Count(*) AS c FROM l7_flow_log GROUP BY time(60)")
    total = sum(row[1] for row in r["values"])
    assert total == N
    base = CFG.base_time_ns // 10**9
    for row in r["values"]:
        assert (row[0] - base) % 60 == 0


def test_select_rows(eng):
    t0 = truth()
    fid = t0[5]["base"]["flow_id"]
    r = eng.query(
        f"SELECT trace_id, request_domain, response_code FROM l7_flow_log "
        f"WHERE flow_id = {fid} LIMIT 10")
    assert len(r["values"]) == 1
    assert r["values"][0][0] == t0[5]["trace_info"]["trace_id"]
    assert r["values"][0][1] == t0[5]["req"]["domain"]
    assert r["values"][0][2] == t0[5]["resp"]["code"]


def test_kg_group(eng):
    r = eng.query(
        "SELECT service_id_1, Count(*) AS c FROM l7_flow_log "
        "GROUP BY service_id_1")
    total = sum(row[1] for row in r["values"])
    assert total == N
    # service ids come from the platform table (1 + ip % 256)
    assert all(row[0] > 0 for row in r["values"])


def test_show_tags(eng):
    r = eng.query("show tags from l7_flow_log")
    names = {row[0] for row in r["values"]}
    assert {"request_domain", "pod_id_0", "response_status",
            "l7_protocol"} <= names


def test_time_range_filter(eng):
    base_s = CFG.base_time_ns // 10**9
    r = eng.query(
        f"SELECT Count(*) AS c FROM l7_flow_log WHERE time >= {base_s} "
        f"AND time <= {base_s + 10}")
    # spans spaced 1ms apart -> ~all within 1s window plus jitter
    assert 0 < r["values"][0][0] <= N


def test_or_clause(eng):
    t0 = truth()
    d0, d1 = t0[0]["req"]["domain"], t0[1]["req"]["domain"]
    r = eng.query(
        f"SELECT Count(*) AS c FROM l7_flow_log WHERE "
        f"(request_domain = '{d0}' OR request_domain = '{d1}')")
    want = sum(1 for t in t0 if t["req"]["domain"] in (d0, d1))
    assert r["values"] == [[want]]


def test_in_list(eng):
    t0 = truth()
    r = eng.query(
        "SELECT Count(*) AS c FROM l7_flow_log WHERE "
        "response_code IN (200, 500)")
    assert r["values"] == [[len(t0)]]
    r2 = eng.query(
        "SELECT Count(*) AS c FROM l7_flow_log WHERE response_code IN (404)")
    assert r2["values"] == [] or r2["values"] == [[0]]


def test_or_with_and(eng):
    t0 = truth()
    d0 = t0[0]["req"]["domain"]
    r = eng.query(
        f"SELECT Count(*) AS c FROM l7_flow_log WHERE server_port = 8080 "
        f"AND (request_domain = '{d0}' OR request_domain = 'nope')")
    want = sum(1 for t in t0 if t["req"]["domain"] == d0)
    assert r["values"] == [[want]]


def test_percentile(eng):
    import numpy as np
    rrts = sorted(t["base"]["head"]["rrt"] for t in truth())
    r = eng.query(
        "SELECT Percentile(response_duration, 50) AS p50 FROM l7_flow_log")
    got = r["values"][0][0]
    med = float(np.quantile(np.array(rrts, dtype=float), 0.5))
    assert abs(got - med) < max(2.0, med * 0.01)


def test_percentile_grouped(eng):
    r = eng.query(
        "SELECT response_status, Percentile(response_duration, 90) AS p "
        "FROM l7_flow_log GROUP BY response_status")
    assert len(r["values"]) == 2
    assert all(row[1] > 0 for row in r["values"])


def test_apdex(eng):
    t0 = truth()
    T = 100000.0
    sat = sum(1 for t in t0 if t["base"]["head"]["rrt"] <= T)
    tol = sum(1 for t in t0 if T < t["base"]["head"]["rrt"] <= 4 * T)
    want = (sat + tol / 2) / len(t0)
    r = eng.query(
        "SELECT Apdex(response_duration, 100000) AS a FROM l7_flow_log")
    assert abs(r["values"][0][0] - want) < 1e-9


def test_slimit(eng):
    r = eng.query(
        "SELECT request_domain, time(60), Count(*) AS c FROM l7_flow_log "
        "GROUP BY request_domain, time(60) SLIMIT 2")
    domains = {row[0] for row in r["values"]}
    assert len(domains) == 2
    # the two kept series are the two biggest domains
    full = eng.query(
        "SELECT request_domain, Count(*) AS c FROM l7_flow_log "
        "GROUP BY request_domain ORDER BY c DESC")
    top2 = {row[0] for row in full["values"][:2]}
    assert domains == top2


def test_application_map_table(eng):
    r = eng.query(
        "SELECT ip_1, Sum(request) AS req FROM application_map "
        "GROUP BY ip_1 ORDER BY req DESC LIMIT 5")
    assert sum(row[1] for row in r["values"]) <= N
    assert len(r["values"]) >= 1
    full = eng.query("SELECT Sum(request) AS r FROM application_map")
    assert full["values"][0][0] == N


def test_attribute_filter(eng):
    t0 = truth()
    val = t0[0]["ext_info"]["attribute_values"][1]
    want = sum(1 for t in t0 if t["ext_info"]["attribute_values"][1] == val)
    r = eng.query(
        f"SELECT Count(*) AS c FROM l7_flow_log WHERE "
        f"attribute.attr_1 = '{val}'")
    assert r["values"] == [[want]]
    # unknown value -> empty; != unknown -> everything
    r2 = eng.query(
        "SELECT Count(*) AS c FROM l7_flow_log WHERE "
        "attribute.attr_1 = 'v9999999'")
    assert r2["values"] == []
    r3 = eng.query(
        f"SELECT Count(*) AS c FROM l7_flow_log WHERE "
        f"attribute.attr_1 != '{val}'")
    assert r3["values"] == [[N - want]]


def test_grouped_percentile_many_groups(eng):
    """Percentile over more than 256 groups (the old cap) — checked
    against direct per-group recomputation from pb truth."""
    import numpy as np
    r = eng.query(
        "SELECT client_port, Percentile(response_duration, 50) AS p50 "
        "FROM l7_flow_log GROUP BY client_port")
    assert len(r["values"]) > 0
    want = {}
    for t in truth():
        want.setdefault(t["base"]["port_src"], []).append(
            t["base"]["head"]["rrt"])
    for port, p50 in r["values"]:
        exp = float(np.quantile(np.array(want[port], dtype=np.float64),
                                0.5))
        assert abs(p50 - exp) < 1e-6, (port, p50, exp)
    assert len(r["values"]) == len(want)


def test_select_attribute_column(eng):
    """attribute.<name> as a SELECT column hydrates per-row values."""
    t0 = gen_span_dict(CFG, 0)
    names = t0["ext_info"]["attribute_names"]
    vals = t0["ext_info"]["attribute_values"]
    name = names[0]
    r = eng.query(f"SELECT request_resource, attribute.{name} "
                  f"FROM l7_flow_log LIMIT 400")
    got = {tuple(row) for row in r["values"]}
    # every truth row appears with its attribute value
    for i in range(5):
        t = gen_span_dict(CFG, i)
        nm = dict(zip(t["ext_info"]["attribute_names"],
                      t["ext_info"]["attribute_values"]))
        assert (t["req"]["resource"], nm[name]) in got


def test_show_descriptions(eng):
    """db_descriptions parity: show tags/metrics carry display name,
    unit and description, generated from the live tag maps for every
    table (reference: querier/db_descriptions 203 data files)."""
    r = eng.query("SHOW tags FROM l7_flow_log")
    assert r["columns"] == ["name", "display_name", "unit", "type",
                            "description"]
    by_name = {v[0]: v for v in r["values"]}
    assert by_name["request_domain"][1] == "Request Domain"
    assert "Host" in by_name["request_domain"][4]
    assert by_name["pod_id_1"][1] == "Server Pod"
    r = eng.query("SHOW metrics FROM l7_flow_log")
    mm = {v[0]: v for v in r["values"]}
    assert mm["response_duration"][2] == "us"
    # rollup tables are covered too
    r = eng.query("SHOW metrics FROM application.1s")
    mm = {v[0]: v for v in r["values"]}
    assert "request" in mm and mm["rrt_max"][2] == "us"
    r = eng.query("SHOW tags FROM application_map.1s")
    names = {v[0] for v in r["values"]}
    assert {"ip_0", "ip_1", "server_port"} <= names


def test_cold_segment_time_pruning():
    """Queries with a time predicate skip cold segments entirely outside
    it (partition pruning) and still return exact results."""
    from deepflow_amd.gen.spans import SpanGenConfig, gen_span_payload
    from deepflow_amd.ingest import L7IngestPipeline
    from deepflow_amd.query import QueryEngine

    base_s = 1_700_000_000
    pipes = L7IngestPipeline(device="cpu", segment_rows=1 << 9,
                             time_base_s=base_s)
    # two 400-span batches an hour apart -> separate segments
    for dt_h in (0, 1):
        cfg = SpanGenConfig(n=400, seed=41 + dt_h, tag_cardinality=50,
                            n_ips=16,
                            base_time_ns=(base_s + dt_h * 3600) * 10**9)
        pipes.ingest_frame_payload(gen_span_payload(cfg))
    while pipes.segments.demote_oldest():
        pass
    cold = pipes.segments.cold
    assert len(cold) >= 1   # hour-1 segment demoted; the tail stays hot
    assert all(getattr(c, "time_max", 0) for c in cold)
    # range covering only the second hour prunes the first segment
    lo = (base_s + 3600) * 10**9
    pruned = pipes.segments.scan_list(time_range=(lo, (1 << 63)))
    full = pipes.segments.scan_list()
    assert len(pruned) < len(full)
    pipes.segments.release_scratch()
    eng = QueryEngine(pipes, device="cpu")
    r = eng.query("SELECT COUNT(1) FROM l7_flow_log "
                  f"WHERE time >= {base_s + 3600}")
    assert r["values"][0][0] == 400
    r = eng.query("SELECT COUNT(1) FROM l7_flow_log")
    assert r["values"][0][0] == 800


def test_hot_segment_time_pruning():
    from deepflow_amd.gen.spans import SpanGenConfig, gen_span_payload
    from deepflow_amd.ingest import L7IngestPipeline
    from deepflow_amd.query import QueryEngine

    base_s = 1_700_000_000
    pipe = L7IngestPipeline(device="cpu", segment_rows=1 << 9,
                            time_base_s=base_s)
    for dt_h in (0, 1):
        cfg = SpanGenConfig(n=400, seed=51 + dt_h, tag_cardinality=50,
                            n_ips=16,
                            base_time_ns=(base_s + dt_h * 3600) * 10**9)
        pipe.ingest_frame_payload(gen_span_payload(cfg))
    assert len(pipe.segments.segments) == 2  # both hot
    lo = (base_s + 3600) * 10**9
    assert len(pipe.segments.scan_list(time_range=(lo, 1 << 63))) == 1
    eng = QueryEngine(pipe, device="cpu")
    r = eng.query("SELECT COUNT(1) FROM l7_flow_log "
                  f"WHERE time >= {base_s + 3600}")
    assert r["values"][0][0] == 400
