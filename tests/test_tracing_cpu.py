"""Tempo + distributed-tracing tests: hand-built multi-hop trace ingested
through the pipeline, fetched by trace_id (string-hash filter), assembled
into a tree via parent-span and syscall joins."""
import pytest
from fastapi.testclient import TestClient

from deepflow_amd.server import DeepflowServer
from deepflow_amd.wire import pb, flow_log, framing

TRACE = "aabbccddeeff00112233445566778899"


def mk_span(span_id, parent="", tap_side=1, svc="front", res="/api/x",
            t0=1_700_000_000_000_000_000, dur_ns=5_000_000,
            syscall_req=0, syscall_resp=0):
    return {
        "base": {
            "start_time": t0,
            "end_time": t0 + dur_ns,
            "flow_id": 42,
            "vtap_id": 1,
            "tap_side": tap_side,
            "head": {"proto": 20, "msg_type": 2, "rrt": dur_ns // 1000},
            "ip_src": 0x0A000001,
            "ip_dst": 0x0A000002,
            "l3_epc_id_src": 1,
            "l3_epc_id_dst": 1,
            "port_src": 40000,
            "port_dst": 8080,
            "protocol": 6,
            "syscall_trace_id_request": syscall_req,
            "syscall_trace_id_response": syscall_resp,
        },
        "req": {"req_type": "GET", "domain": "svc", "resource": res,
                "endpoint": res},
        "resp": {"status": 0, "code": 200},
        "trace_info": {"trace_id": TRACE, "span_id": span_id,
                       "parent_span_id": parent},
        "ext_info": {"service_name": svc},
    }


@pytest.fixture(scope="module")
def server():
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12)
    spans = [
        mk_span("s-root", tap_side=1, svc="front", syscall_resp=777),
        mk_span("s-mid", parent="s-root", tap_side=0, svc="mid",
                syscall_req=777, syscall_resp=888),
        # third hop: no parent_span_id, linked via syscall ids
        mk_span("s-leaf", tap_side=0, svc="back", syscall_req=888),
        # unrelated trace
        dict(mk_span("other"), trace_info={"trace_id": "ff" * 16,
                                           "span_id": "zz"}),
    ]
    payload = framing.pack_records(
        [pb.encode(s, flow_log.APP_PROTO_LOGS_DATA) for s in spans])
    srv.receiver.handle_frame(framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG), payload))
    return srv


def test_trace_id_filter(server):
    r = server.engine.query(
        f"SELECT span_id, service_name FROM l7_flow_log "
        f"WHERE trace_id = '{TRACE}' LIMIT 100")
    ids = sorted(v[0] for v in r["values"])
    assert ids == ["s-leaf", "s-mid", "s-root"]


def test_tempo_api(server):
    client = TestClient(server.app)
    r = client.get(f"/api/traces/{TRACE}")
    body = r.json()
    assert body["spanCount"] == 3
    services = {b["resource"]["attributes"][0]["value"]["stringValue"]
                for b in body["batches"]}
    assert services == {"front", "mid", "back"}
    assert client.get("/api/echo").status_code == 200


def test_trace_tree(server):
    client = TestClient(server.app)
    body = client.get(f"/v1/tracing/{TRACE}").json()
    assert body["span_count"] == 3
    nodes = {n["span_id"]: n for n in body["spans"]}
    assert nodes["s-root"]["parent_index"] is None
    assert body["spans"][nodes["s-mid"]["parent_index"]]["span_id"] == "s-root"
    # leaf linked via syscall_trace_id join (no parent_span_id on wire)
    assert body["spans"][nodes["s-leaf"]["parent_index"]]["span_id"] == "s-mid"
    assert len(body["roots"]) == 1


def test_trace_tree_table(server):
    from fastapi.testclient import TestClient
    client = TestClient(server.app)
    client.get(f"/v1/tracing/{TRACE}")
    r = client.post("/v1/query/", json={
        "sql": "SELECT trace_id, span_count, max_depth FROM trace_tree "
               "LIMIT 10"})
    vals = r.json()["result"]["values"]
    assert [TRACE, 3, 2] in vals
