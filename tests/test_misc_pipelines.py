"""Events, app logs, OTLP exporter, MCP server tests."""
import pytest
from fastapi.testclient import TestClient

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload, gen_span_dict
from deepflow_amd.server import DeepflowServer
from deepflow_amd.store.kg import KgInfo
from deepflow_amd.wire import pb, framing
from deepflow_amd.ingest.event_pipeline import PROC_EVENT

CFG = SpanGenConfig(n=60, seed=88, tag_cardinality=20, n_ips=32,
                    n_services=4)


@pytest.fixture(scope="module")
def server():
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12,
                         time_base_s=CFG.base_time_ns // 10**9)
    srv.receiver.handle_frame(framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG),
        gen_span_payload(CFG)))
    return srv


def test_resource_events_from_platform(server):
    server.controller.update_platform(
        {(3, 0x0A0000AA): KgInfo(pod_id=55)})
    assert any(e["resource_type"] == "pod" and e["resource_id"] == 55
               for e in server.events.resource_events)
    r = server.engine.query("SELECT resource_type, Count(*) AS c FROM event "
                            "GROUP BY resource_type")
    assert r["values"][0][1] >= 1


def test_proc_events(server):
    rec = pb.encode({
        "pid": 1234, "thread_id": 1, "process_kname": b"nginx\x00",
        "start_time": 5_000_000_000, "end_time": 5_002_000_000,
        "event_type": 0, "pod_id": 9,
        "io_event_data": {"bytes_count": 4096, "operation": 1,
                          "latency": 250, "filename": b"/var/log/x\x00"},
    }, PROC_EVENT)
    frame = framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_PROC_EVENT),
        framing.pack_records([rec]))
    assert server.receiver.handle_frame(frame)
    r = server.engine.query(
        "SELECT filename, Sum(bytes_count) AS b FROM perf_event "
        "GROUP BY filename")
    assert ["/var/log/x", 4096] in r["values"]


def test_app_logs(server):
    lines = b"\n".join([
        b"1700000100 ERROR checkout failed to connect to db",
        b"1700000101 info checkout request ok",
        b"free form line",
    ])
    frame = framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_APPLICATION_LOG,
                            agent_id=6), lines)
    assert server.receiver.handle_frame(frame)
    errs = server.applogs.search(severity_max=3)
    assert any("failed to connect" in r["body"] for r in errs)
    r = server.engine.query(
        "SELECT severity, Count(*) AS c FROM application_log "
        "GROUP BY severity")
    assert sum(v[1] for v in r["values"]) == 3


def test_otlp_export_roundtrip(server):
    blob = server.exporter.export_where(limit=100)
    # feed our own exporter output back through the OTLP importer
    from deepflow_amd.ingest.otel import otlp_to_l7_payload
    payload = otlp_to_l7_payload(blob, compressed=False)
    from deepflow_amd.wire import flow_log
    recs = list(framing.iter_records(payload))
    assert len(recs) == 60
    d = pb.decode(recs[0], flow_log.APP_PROTO_LOGS_DATA)
    assert d["trace_info"]["trace_id"]
    client = TestClient(server.app)
    resp = client.get("/v1/export/otlp", params={"limit": 10})
    assert resp.status_code == 200 and len(resp.content) > 100


def test_mcp(server):
    client = TestClient(server.app)
    r = client.post("/mcp", json={"jsonrpc": "2.0", "id": 1,
                                  "method": "initialize", "params": {}})
    assert r.json()["result"]["serverInfo"]["name"] == "deepflow-amd-mcp"
    r2 = client.post("/mcp", json={"jsonrpc": "2.0", "id": 2,
                                   "method": "tools/list"})
    names = {t["name"] for t in r2.json()["result"]["tools"]}
    assert names == {"profile_analysis", "query"}
    r3 = client.post("/mcp", json={
        "jsonrpc": "2.0", "id": 3, "method": "tools/call",
        "params": {"name": "query",
                   "arguments": {"sql":
                                 "SELECT Count(*) AS c FROM l7_flow_log"}}})
    assert "60" in r3.json()["result"]["content"][0]["text"]
