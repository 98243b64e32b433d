"""OTLP logs + metrics ingest: LogsData -> application_log rows,
MetricsData (gauge/sum/histogram) -> ID-encoded prometheus samples,
queryable via PromQL. Wire bytes built with the schema codec
(opentelemetry-proto logs/v1 + metrics/v1 field numbers)."""
from fastapi.testclient import TestClient

from deepflow_amd.server import DeepflowServer
from deepflow_amd.wire import pb, otlp


def _kv(k, v):
    return {"key": k, "value": {"string_value": v}}


def logs_blob():
    return pb.encode({
        "resource_logs": [{
            "resource": {"attributes": [_kv("service.name", "checkout")]},
            "scope_logs": [{"log_records": [
                {"time_unix_nano": 5 * 10**9, "severity_number": 17,
                 "severity_text": "ERROR",
                 "body": {"string_value": "payment failed"},
                 "attributes": [_kv("order", "o-17")],
                 "trace_id": bytes.fromhex("aa" * 16)},
                {"time_unix_nano": 6 * 10**9, "severity_number": 9,
                 "body": {"string_value": "ok"}},
            ]}],
        }],
    }, otlp.LOGS_DATA)


def metrics_blob():
    return pb.encode({
        "resource_metrics": [{
            "resource": {"attributes": [_kv("service.name", "checkout")]},
            "scope_metrics": [{"metrics": [
                {"name": "queue_depth", "gauge": {"data_points": [
                    {"time_unix_nano": 10**9, "as_int": 42,
                     "attributes": [_kv("queue", "q0")]}]}},
                {"name": "requests_total", "sum": {
                    "is_monotonic": 1, "aggregation_temporality": 2,
                    "data_points": [
                        {"time_unix_nano": 10**9, "as_double": 100.0},
                        {"time_unix_nano": 61 * 10**9, "as_double": 160.0},
                    ]}},
                {"name": "latency", "histogram": {"data_points": [
                    {"time_unix_nano": 10**9, "count": 10, "sum": 2.5,
                     "bucket_counts": [6, 3, 1],
                     "explicit_bounds": [0.1, 0.5]}]}},
            ]}],
        }],
    }, otlp.METRICS_DATA)


def test_otlp_logs_roundtrip():
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12)
    client = TestClient(srv.app)
    r = client.post("/otlp/v1/logs", content=logs_blob())
    assert r.json()["accepted"] == 2
    rows = srv.applogs.rows
    err = [x for x in rows if x["severity"] == 3]
    assert len(err) == 1
    assert err[0]["body"] == "payment failed"
    assert err[0]["app_service"] == "checkout"
    assert err[0]["trace_id"] == "aa" * 16
    assert err[0]["attr.order"] == "o-17"
    # queryable through the application_log table
    q = srv.engine.query(
        "SELECT body FROM application_log WHERE severity = 3 LIMIT 10")
    assert q["values"] == [["payment failed"]]


def test_otlp_metrics_to_promql():
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12)
    client = TestClient(srv.app)
    r = client.post("/otlp/v1/metrics", content=metrics_blob())
    # 1 gauge + 2 sum + (count+sum+3 buckets) = 8 samples
    assert r.json()["accepted"] == 8
    assert srv.prom.metric_names.intern("queue_depth") is not None
    # rate over the monotonic sum: (160-100)/60 = 1/s
    res = client.get("/prom/api/v1/query", params={
        "query": "rate(requests_total[2m])", "time": "61"}).json()
    vals = [float(s["value"][1]) for s in res["data"]["result"]]
    assert any(abs(v - 1.0) < 0.05 for v in vals)
    # histogram expanded to prometheus convention
    res2 = client.get("/prom/api/v1/query", params={
        "query": "latency_bucket", "time": "1"}).json()
    les = {s["metric"].get("le"): float(s["value"][1])
           for s in res2["data"]["result"]}
    assert les == {"0.1": 6.0, "0.5": 9.0, "+Inf": 10.0}


def test_histogram_quantile_and_topk():
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12)
    client = TestClient(srv.app)
    client.post("/otlp/v1/metrics", content=metrics_blob())
    r = client.get("/prom/api/v1/query", params={
        "query": "histogram_quantile(0.5, latency_bucket)",
        "time": "1"}).json()
    # buckets: 6 <= 0.1, cum 9 <= 0.5, 10 total; p50 target=5 inside
    # the first bucket -> 0.1 * 5/6
    v = float(r["data"]["result"][0]["value"][1])
    assert abs(v - 0.1 * 5 / 6) < 1e-9, v
    r2 = client.get("/prom/api/v1/query", params={
        "query": "topk(1, latency_bucket)", "time": "1"}).json()
    assert float(r2["data"]["result"][0]["value"][1]) == 10.0
    assert r2["data"]["result"][0]["metric"]["le"] == "+Inf"
