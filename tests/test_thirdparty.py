"""SkyWalking / Datadog trace adapter tests."""
import json

from fastapi.testclient import TestClient

from deepflow_amd.ingest.thirdparty import THIRD_PARTY_TRACE, TracingAdapter
from deepflow_amd.server import DeepflowServer
from deepflow_amd.wire import pb, framing

SW_SEGMENT = {
    "traceId": "sw-trace-0001",
    "traceSegmentId": "seg-1",
    "service": "billing",
    "spans": [
        {"spanId": 0, "parentSpanId": -1, "operationName": "/pay",
         "startTime": 1700000000000, "endTime": 1700000000120,
         "spanType": "Entry", "spanLayer": "Http", "isError": False,
         "tags": [{"key": "http.method", "value": "POST"},
                  {"key": "http.status_code", "value": "200"}]},
        {"spanId": 1, "parentSpanId": 0, "operationName": "SELECT pay",
         "startTime": 1700000000010, "endTime": 1700000000050,
         "spanType": "Exit", "spanLayer": "Database", "isError": True,
         "peer": "db:3306", "tags": []},
    ],
}

DD_TRACES = [[
    {"trace_id": 0xABC, "span_id": 0x1, "parent_id": 0,
     "name": "web.request", "resource": "GET /cart", "service": "cart",
     "start": 1700000001000000000, "duration": 5000000, "type": "web",
     "error": 0, "meta": {"http.method": "GET", "http.status_code": "200"}},
]]


def _server():
    return DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                          dict_capacity=1 << 12, time_base_s=1_700_000_000)


def test_skywalking_frame():
    srv = _server()
    tpt = pb.encode({"data": json.dumps([SW_SEGMENT]).encode(),
                     "uri": "/v3/segments"}, THIRD_PARTY_TRACE)
    frame = framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_SKYWALKING), tpt)
    assert srv.receiver.handle_frame(frame)
    assert srv.l7.stats.spans_in == 2
    r = srv.engine.query(
        "SELECT span_id, service_name FROM l7_flow_log "
        "WHERE trace_id = 'sw-trace-0001' LIMIT 10")
    assert len(r["values"]) == 2
    assert all(v[1] == "billing" for v in r["values"])
    tree = srv.tracer.assemble("sw-trace-0001")
    assert tree["span_count"] == 2 and len(tree["roots"]) == 1


def test_datadog_frame():
    srv = _server()
    tpt = pb.encode({"data": json.dumps(DD_TRACES).encode()},
                    THIRD_PARTY_TRACE)
    frame = framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_DATADOG), tpt)
    assert srv.receiver.handle_frame(frame)
    assert srv.l7.stats.spans_in == 1
    r = srv.engine.query(
        "SELECT request_resource, service_name FROM l7_flow_log "
        "WHERE response_code = 200 LIMIT 5")
    assert ["GET /cart", "cart"] in r["values"]


def test_tracing_adapter_pull():
    srv = _server()
    adapter = TracingAdapter(
        lambda payload: srv.l7.ingest_frame_payload(payload))
    n = adapter.import_skywalking_segments([SW_SEGMENT])
    assert n == 2
    assert srv.l7.stats.spans_in == 2
