"""All-in-one server test (BASELINE config #1 shape): agent-side framed
stream over loopback TCP -> receiver -> pipeline -> HTTP query."""
import socket
import time

import pytest
from fastapi.testclient import TestClient

from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.server import DeepflowServer
from deepflow_amd.wire import framing

N = 1000
CFG = SpanGenConfig(n=N, seed=9, tag_cardinality=100, n_ips=64,
                    n_services=8, n_resources=20)


@pytest.fixture(scope="module")
def server():
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 11,
                         dict_capacity=1 << 13,
                         time_base_s=CFG.base_time_ns // 10**9,
                         platform_cfg=CFG)
    srv.start()
    yield srv
    srv.stop()


def test_tcp_ingest_and_query(server):
    payload = gen_span_payload(CFG)
    hdr = framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG, agent_id=3)
    frame = framing.encode_frame(hdr, payload)
    s = socket.create_connection(("127.0.0.1", server.receiver.tcp_port),
                                 timeout=5)
    # send in two chunks to exercise reassembly
    s.sendall(frame[:1000])
    time.sleep(0.05)
    s.sendall(frame[1000:])
    s.close()
    deadline = time.time() + 20
    while time.time() < deadline and server.l7.stats.spans_in < N:
        time.sleep(0.1)
    assert server.l7.stats.spans_in == N

    client = TestClient(server.app)
    r = client.post("/v1/query/", data={"sql":
        "SELECT Count(*) AS cnt FROM l7_flow_log"})
    assert r.status_code == 200
    body = r.json()
    assert body["OPT_STATUS"] == "SUCCESS"
    assert body["result"]["values"] == [[N]]

    r2 = client.post("/v1/query/", json={
        "sql": "SELECT request_domain, Count(*) AS c FROM l7_flow_log "
               "GROUP BY request_domain ORDER BY c DESC LIMIT 3"})
    assert r2.status_code == 200
    vals = r2.json()["result"]["values"]
    assert len(vals) == 3
    assert all(v[0].startswith("svc-") for v in vals)

    # receiver status accounting
    st = server.receiver.status[(3, framing.MSG_PROTOCOLLOG)]
    assert st.frames == 1 and st.bytes == len(frame)


def test_zstd_frame(server):
    import ctypes as ct
    import numpy as np
    from deepflow_amd.ops import native
    before = server.l7.stats.spans_in
    cfg2 = SpanGenConfig(n=50, seed=77, tag_cardinality=100, n_ips=64)
    payload = gen_span_payload(cfg2)
    lib = native.cpu()
    src = np.frombuffer(payload, dtype=np.uint8)
    dst = np.zeros(len(payload) * 2 + 1024, dtype=np.uint8)
    n = lib.df_zstd_compress(src.ctypes.data, len(src), dst.ctypes.data,
                             len(dst), 3)
    assert n > 0
    hdr = framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG, agent_id=4,
                              encoder=framing.ENCODER_ZSTD)
    frame = framing.encode_frame(hdr, dst[:n].tobytes())
    assert server.receiver.handle_frame(frame)
    assert server.l7.stats.spans_in == before + 50


def test_health_and_stats(server):
    client = TestClient(server.app)
    assert client.get("/v1/health").json()["status"] == "ok"
    stats = client.get("/v1/stats").json()
    names = {s["name"] for s in stats}
    assert "ingester.receiver" in names


def test_l4_ingest_and_query(server):
    from deepflow_amd.gen import FlowGenConfig
    from deepflow_amd.gen.flows import gen_flow_payload, gen_flow_dict
    fcfg = FlowGenConfig(n=80, seed=19, n_ips=32, n_epcs=8)
    payload = gen_flow_payload(fcfg)
    hdr = framing.FrameHeader(msg_type=framing.MSG_TAGGEDFLOW, agent_id=5)
    assert server.receiver.handle_frame(framing.encode_frame(hdr, payload))
    assert server.l4.stats.flows_in == 80
    client = TestClient(server.app)
    r = client.post("/v1/query/", json={
        "sql": "SELECT Count(*) AS c, Sum(byte_tx) AS b FROM l4_flow_log"})
    body = r.json()
    assert body["OPT_STATUS"] == "SUCCESS", body
    want_b = sum(gen_flow_dict(fcfg, i)["flow"]["metrics_peer_src"]["byte_count"]
                 for i in range(80))
    assert body["result"]["values"] == [[80, want_b]]
    # select with string column
    r2 = client.post("/v1/query/", json={
        "sql": "SELECT flow_id, server_port, close_type FROM l4_flow_log "
               "WHERE protocol = 6 LIMIT 5"})
    assert len(r2.json()["result"]["values"]) == 5


def test_metrics_tables(server):
    client = TestClient(server.app)
    r = client.post("/v1/query/", json={
        "sql": "SELECT time(60), Sum(request) AS req FROM application "
               "GROUP BY time(60)"})
    body = r.json()
    assert body["OPT_STATUS"] == "SUCCESS", body
    total = sum(v[1] for v in body["result"]["values"])
    assert total == server.l7.stats.spans_in
    r2 = client.post("/v1/query/", json={
        "sql": "SELECT Sum(byte_tx) AS b FROM network"})
    assert r2.json()["OPT_STATUS"] == "SUCCESS"
    assert r2.json()["result"]["values"][0][0] > 0


def test_concurrent_ingest_and_query():
    """Queries serialize against ingest via the shared engine lock —
    hammer both concurrently and check invariants hold."""
    import threading
    from deepflow_amd.server import DeepflowServer
    from deepflow_amd.wire import pb, flow_log, framing
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 9,
                         dict_capacity=1 << 12)

    def span(i):
        return {"base": {"start_time": 10**18 + i, "end_time": 10**18 + i,
                         "flow_id": i, "vtap_id": 1, "tap_side": 1,
                         "head": {"proto": 20, "msg_type": 2, "rrt": 5},
                         "ip_src": 1, "ip_dst": 2, "port_src": 9,
                         "port_dst": 80, "protocol": 6},
                "req": {"req_type": "GET", "domain": "d",
                        "resource": f"/r{i % 7}", "endpoint": "e"}}

    stop = threading.Event()
    errors = []

    def ingester():
        i = 0
        while not stop.is_set():
            recs = [pb.encode(span(i * 50 + j),
                              flow_log.APP_PROTO_LOGS_DATA)
                    for j in range(50)]
            srv.receiver.handle_frame(framing.encode_frame(
                framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG),
                framing.pack_records(recs)))
            i += 1

    def querier():
        while not stop.is_set():
            try:
                r = srv.engine.query(
                    "SELECT request_resource, Count(*) AS c "
                    "FROM l7_flow_log GROUP BY request_resource")
                total = sum(v[1] for v in r["values"])
                if total % 50 != 0:  # partial batch visible mid-ingest
                    errors.append(f"torn read: {total}")
            except Exception as e:  # noqa: BLE001
                errors.append(repr(e))

    threads = [threading.Thread(target=ingester),
               threading.Thread(target=querier),
               threading.Thread(target=querier)]
    for t in threads:
        t.start()
    import time
    time.sleep(2.0)
    stop.set()
    for t in threads:
        t.join(timeout=10)
    assert not errors, errors[:3]
    assert srv.l7.stats.spans_in > 0
