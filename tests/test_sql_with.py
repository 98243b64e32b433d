"""WITH / derived-subquery DF-SQL tests (CHEngine WITH-clause parity):
inner query runs on the GPU/CPU segment scan, outer re-aggregates the
materialized rows."""
import pytest

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.ingest import L7IngestPipeline
from deepflow_amd.query.engine import QueryEngine, split_with

CFG = SpanGenConfig(n=3000, seed=7, tag_cardinality=40, n_attrs=2,
                    n_ips=64, n_services=4, n_resources=10)


@pytest.fixture(scope="module")
def engine():
    pipe = L7IngestPipeline(device="cpu", segment_rows=1 << 12,
                            dict_capacity=1 << 14,
                            time_base_s=CFG.base_time_ns // 10**9)
    pipe.ingest_frame_payload(gen_span_payload(CFG))
    return QueryEngine(pipe, device="cpu")


def test_split_with():
    ctes, main = split_with(
        "WITH a AS (SELECT x FROM t), b AS (SELECT y FROM a) "
        "SELECT * FROM b")
    assert [c[0] for c in ctes] == ["a", "b"]
    assert ctes[0][1] == "SELECT x FROM t"
    assert main.strip() == "SELECT * FROM b"


def test_with_reaggregate(engine):
    # per-(proto, status) counts -> total per proto; must equal the direct
    # per-proto counts
    r = engine.query(
        "WITH per AS (SELECT l7_protocol, response_status, Count(*) AS c "
        "FROM l7_flow_log GROUP BY l7_protocol, response_status) "
        "SELECT l7_protocol, Sum(c) AS total FROM per GROUP BY l7_protocol")
    direct = engine.query(
        "SELECT l7_protocol, Count(*) AS total FROM l7_flow_log "
        "GROUP BY l7_protocol")
    assert sorted(map(tuple, r["values"])) == \
        sorted(map(tuple, direct["values"]))


def test_derived_table(engine):
    r = engine.query(
        "SELECT Count(*) AS n FROM (SELECT request_resource, Count(*) AS c "
        "FROM l7_flow_log GROUP BY request_resource) t WHERE c > 0")
    inner = engine.query(
        "SELECT request_resource, Count(*) AS c FROM l7_flow_log "
        "GROUP BY request_resource")
    assert r["values"][0][0] == len(inner["values"])


def test_with_string_filter(engine):
    # 'raw' hydrate: string filter on a CTE column
    r = engine.query(
        "WITH per AS (SELECT l7_protocol, Count(*) AS c FROM l7_flow_log "
        "GROUP BY l7_protocol) "
        "SELECT c FROM per WHERE l7_protocol = 'HTTP'")
    direct = engine.query(
        "SELECT Count(*) AS c FROM l7_flow_log WHERE l7_protocol = 'HTTP'")
    assert r["values"][0][0] == direct["values"][0][0]


def test_with_chained_ctes(engine):
    r = engine.query(
        "WITH a AS (SELECT l7_protocol, response_status, Count(*) AS c "
        "FROM l7_flow_log GROUP BY l7_protocol, response_status), "
        "b AS (SELECT l7_protocol, Sum(c) AS s FROM a GROUP BY l7_protocol) "
        "SELECT Sum(s) AS grand FROM b")
    total = engine.query("SELECT Count(*) AS n FROM l7_flow_log")
    assert r["values"][0][0] == total["values"][0][0]


def test_row_table_or_group(engine):
    # OR clause against a derived table exercises the CNF path in _run_rows
    r = engine.query(
        "SELECT Sum(c) AS s FROM (SELECT l7_protocol, Count(*) AS c "
        "FROM l7_flow_log GROUP BY l7_protocol) t "
        "WHERE (l7_protocol = 'HTTP' OR l7_protocol = 'DNS')")
    direct = engine.query(
        "SELECT Count(*) AS c FROM l7_flow_log "
        "WHERE (l7_protocol = 'HTTP' OR l7_protocol = 'DNS')")
    assert r["values"][0][0] == direct["values"][0][0]


def test_having(engine):
    r = engine.query(
        "SELECT request_resource, Count(*) AS c FROM l7_flow_log "
        "GROUP BY request_resource HAVING c >= 200 ORDER BY c DESC")
    assert r["values"], "expected some groups over the threshold"
    assert all(row[1] >= 200 for row in r["values"])
    full = engine.query(
        "SELECT request_resource, Count(*) AS c FROM l7_flow_log "
        "GROUP BY request_resource")
    want = sorted([v for _, v in map(tuple, full["values"]) if v >= 200],
                  reverse=True)
    assert [row[1] for row in r["values"]] == want
