"""OTLP ingest + controller-lite tests."""
import zlib

import pytest
from fastapi.testclient import TestClient

from deepflow_amd.server import DeepflowServer
from deepflow_amd.store.kg import KgInfo
from deepflow_amd.wire import pb, otlp, framing


def mk_traces_data():
    def kv(k, v):
        if isinstance(v, int):
            return {"key": k, "value": {"int_value": v}}
        return {"key": k, "value": {"string_value": v}}

    spans = []
    for i in range(8):
        spans.append({
            "trace_id": bytes.fromhex("%032x" % (0xABC000 + i)),
            "span_id": bytes.fromhex("%016x" % (0xDEF000 + i)),
            "name": "GET /checkout",
            "kind": otlp.SPAN_KIND_CLIENT,
            "start_time_unix_nano": 1_700_000_000_000_000_000 + i * 10**6,
            "end_time_unix_nano": 1_700_000_000_000_000_000 + i * 10**6 + 3 * 10**6,
            "attributes": [
                kv("http.method", "GET"),
                kv("http.target", "/checkout"),
                kv("http.host", "shop.local"),
                kv("http.status_code", 200 if i % 4 else 503),
                kv("user.tier", "gold"),
            ],
            "status": {"code": 2 if i % 4 == 0 else 1},
        })
    td = {
        "resource_spans": [{
            "resource": {"attributes": [kv("service.name", "checkout-svc")]},
            "scope_spans": [{"scope": {"name": "otel-sdk"}, "spans": spans}],
        }]
    }
    return pb.encode(td, otlp.TRACES_DATA)


@pytest.fixture(scope="module")
def server():
    return DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                          dict_capacity=1 << 12,
                          time_base_s=1_700_000_000)


def test_otlp_ingest(server):
    blob = zlib.compress(mk_traces_data())
    hdr = framing.FrameHeader(msg_type=framing.MSG_OPENTELEMETRY, agent_id=9)
    assert server.receiver.handle_frame(framing.encode_frame(hdr, blob))
    assert server.l7.stats.spans_in == 8
    r = server.engine.query(
        "SELECT request_domain, service_name, Count(*) AS c FROM l7_flow_log "
        "WHERE request_resource = '/checkout' GROUP BY request_domain, "
        "service_name")
    assert r["values"] == [["shop.local", "checkout-svc", 8]]
    # otel error status mapped to server error (503)
    r2 = server.engine.query(
        "SELECT Count(*) AS c FROM l7_flow_log WHERE response_code = 503")
    assert r2["values"] == [[2]]
    # custom attribute interned
    r3 = server.engine.query("show tag attribute.x values from l7_flow_log")
    assert ["gold"] in r3["values"]


def test_controller_sync(server):
    client = TestClient(server.app)
    r = client.post("/v1/sync/", json={"agent_id": 11, "hostname": "nodeA",
                                       "config_version": 0,
                                       "platform_version": 0})
    body = r.json()
    assert body["status"] == "ok"
    assert "config" in body and body["config"]["sync_interval"] == 10
    assert "platform" in body
    # second sync with current versions -> no payload push
    r2 = client.post("/v1/sync/", json={
        "agent_id": 11, "config_version": body["config_version"],
        "platform_version": body["platform_version"]})
    b2 = r2.json()
    assert "config" not in b2 and "platform" not in b2
    agents = client.get("/v1/agents/").json()
    assert any(a["agent_id"] == 11 and a["alive"] for a in agents)


def test_controller_platform_push(server):
    server.controller.update_platform(
        {(9, 0x0A00FFFF): KgInfo(pod_id=777, service_id=12)},
        names={"pod_map": {777: "pod-qa-1"}})
    assert server.kg.lookup(9, 0x0A00FFFF).pod_id == 777
    assert server.controller.lookup_name("pod_map", 777) == "pod-qa-1"
    client = TestClient(server.app)
    r = client.post("/v1/sync/", json={"agent_id": 11, "config_version": 99,
                                       "platform_version": 0})
    assert any(p["pod_id"] == 777 for p in r.json()["platform"])


def test_config_push_version_bump(server):
    client = TestClient(server.app)
    v0 = server.controller.config_version
    r = client.post("/v1/agent-group-config/default",
                    json={"throttle_per_second": 1000})
    assert r.json()["config_version"] == v0 + 1
    r2 = client.post("/v1/sync/", json={"agent_id": 11,
                                        "config_version": v0,
                                        "platform_version": 999})
    assert r2.json()["config"]["throttle_per_second"] == 1000


def test_native_otlp_converter_matches_python():
    """C++ df_otlp_to_l7 output is byte-identical to the Python
    converter across span shapes (http/grpc/db/plain + statuses)."""
    import numpy as np
    from deepflow_amd.wire import pb, otlp
    from deepflow_amd.ingest.otel import otlp_to_l7_payload
    from deepflow_amd.ops import native

    def kv(k, v):
        return {"key": k, "value": {"string_value": v}}

    spans = [
        {"trace_id": bytes.fromhex("aa" * 16), "span_id": b"\x01" * 8,
         "name": "GET /x", "kind": 3,
         "start_time_unix_nano": 10**18,
         "end_time_unix_nano": 10**18 + 5 * 10**6,
         "attributes": [kv("http.method", "GET"), kv("http.target", "/x"),
                        kv("http.host", "h"), kv("team", "core"),
                        {"key": "http.status_code",
                         "value": {"int_value": 503}},
                        {"key": "retries", "value": {"int_value": 2}},
                        {"key": "cached", "value": {"bool_value": True}}],
         "status": {"code": 2}},
        {"trace_id": bytes.fromhex("bb" * 16), "span_id": b"\x02" * 8,
         "parent_span_id": b"\x03" * 8, "name": "svc.Api/Do", "kind": 2,
         "start_time_unix_nano": 10**18,
         "end_time_unix_nano": 10**18 + 10**6,
         "attributes": [kv("rpc.system", "grpc"), kv("rpc.method", "Do"),
                        kv("rpc.service", "svc.Api")]},
        {"name": "lonely-span", "kind": 0,
         "start_time_unix_nano": 5, "end_time_unix_nano": 9},
        {"name": "q", "kind": 2,
         "start_time_unix_nano": 1, "end_time_unix_nano": 2,
         "attributes": [kv("db.system", "redis"), kv("db.operation", "GET"),
                        kv("db.name", "cache"),
                        kv("db.statement", "GET k")]},
    ]
    blob = pb.encode({"resource_spans": [{
        "resource": {"attributes": [kv("service.name", "checkout")]},
        "scope_spans": [{"spans": spans}]}]}, otlp.TRACES_DATA)
    want = otlp_to_l7_payload(blob)
    lib = native.cpu()
    src = np.frombuffer(blob, dtype=np.uint8)
    need = int(lib.df_otlp_to_l7(src.ctypes.data, len(src), None, 0))
    dst = np.zeros(need, dtype=np.uint8)
    got = int(lib.df_otlp_to_l7(src.ctypes.data, len(src),
                                dst.ctypes.data, need))
    assert got == need
    assert dst.tobytes() == want
    # malformed input refuses cleanly
    bad = np.frombuffer(b"\xff\xff\xff\xff\x02", dtype=np.uint8)
    assert lib.df_otlp_to_l7(bad.ctypes.data, len(bad), None, 0) <= 0
