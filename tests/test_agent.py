"""C++ agent core tests: packet -> flow -> L7 parse -> wire records, plus
the all-in-one loop (BASELINE config #1: agent -> ingester -> query)."""
import time

import pytest
from fastapi.testclient import TestClient

from deepflow_amd.agent import Agent
from deepflow_amd.agent.packets import (http_session, dns_session,
                                        redis_session)
from deepflow_amd.server import DeepflowServer
from deepflow_amd.wire import pb, flow_log, metric, framing

CLIENT = 0x0A000001
SERVER = 0x0A000002


@pytest.fixture()
def agent():
    a = Agent(vtap_id=7)
    a.add_cidr(0x0A000000, 8, epc=42)
    yield a
    a.close()


def _decode(payload, schema):
    return [pb.decode(r, schema) for r in framing.iter_records(payload)]


def test_http_flow(agent):
    for frame, ts in http_session(CLIENT, SERVER, code=200):
        assert agent.packet(frame, ts) == 0
    agent.tick(10**9 * 100)
    l7 = _decode(agent.drain(1), flow_log.APP_PROTO_LOGS_DATA)
    assert len(l7) == 1
    rec = l7[0]
    assert rec["base"]["vtap_id"] == 7
    assert rec["base"]["ip_src"] == CLIENT
    assert rec["base"]["l3_epc_id_src"] == 42
    assert rec["base"]["head"]["proto"] == 20
    assert rec["req"]["req_type"] == "GET"
    assert rec["req"]["domain"] == "svc.example.com"
    assert rec["req"]["resource"] == "/api/x"
    assert rec["resp"]["code"] == 200
    assert rec["base"]["head"]["rrt"] == 4000  # 5ms->9ms

    l4 = _decode(agent.drain(0), flow_log.TAGGED_FLOW)
    assert len(l4) == 1
    f = l4[0]["flow"]
    assert f["close_type"] == 1  # both FINs
    assert f["metrics_peer_src"]["packet_count"] == 4
    assert f["metrics_peer_dst"]["packet_count"] == 3
    perf = f["perf_stats"]
    assert perf["l7_protocol"] == 20
    assert perf["tcp"]["rtt"] == 2000  # SYN->SYNACK 2ms
    assert perf["l7"]["request_count"] == 1
    assert perf["l7"]["response_count"] == 1

    docs = _decode(agent.drain(2), metric.DOCUMENT)
    assert len(docs) >= 1
    d = docs[0]
    assert d["meter"]["app"]["traffic"]["request"] == 1
    assert d["tag"]["field"]["server_port"] == 8080


def test_http_error_status(agent):
    for frame, ts in http_session(CLIENT, SERVER, sport=50000, code=500):
        agent.packet(frame, ts)
    agent.tick(10**9 * 100)
    l7 = _decode(agent.drain(1), flow_log.APP_PROTO_LOGS_DATA)
    assert l7[0]["resp"]["status"] == 3
    assert l7[0]["resp"]["code"] == 500


def test_dns(agent):
    for frame, ts in dns_session(CLIENT, SERVER, qname="db.svc.local"):
        agent.packet(frame, ts)
    agent.tick(10**9 * 100)
    l7 = _decode(agent.drain(1), flow_log.APP_PROTO_LOGS_DATA)
    assert len(l7) == 1
    assert l7[0]["base"]["head"]["proto"] == 120
    assert l7[0]["req"]["domain"] == "db.svc.local"


def test_redis(agent):
    for frame, ts in redis_session(CLIENT, SERVER):
        agent.packet(frame, ts)
    agent.tick(10**9 * 100)
    l7 = _decode(agent.drain(1), flow_log.APP_PROTO_LOGS_DATA)
    assert len(l7) == 1
    assert l7[0]["base"]["head"]["proto"] == 80
    assert l7[0]["req"]["req_type"] == "GET"
    assert l7[0]["req"]["resource"] == "mykey"


def test_agent_to_server_end_to_end():
    """BASELINE config #1: agent -> TCP -> ingester -> SQL query."""
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12, time_base_s=0)
    srv.start()
    try:
        a = Agent(vtap_id=3, server=("127.0.0.1", srv.receiver.tcp_port))
        a.add_cidr(0x0A000000, 8, epc=5)
        t0 = 10**9
        for i in range(20):
            for frame, ts in http_session(CLIENT + i, SERVER,
                                          sport=40000 + i,
                                          path=f"/api/item/{i % 4}",
                                          t0=t0 + i * 10**7):
                a.packet(frame, ts)
        sent = a.flush_to_server(10**9 * 100)
        assert sent == 4  # l4 + l7 + doc + dfstats frames
        deadline = time.time() + 20
        while time.time() < deadline and srv.l7.stats.spans_in < 20:
            time.sleep(0.1)
        assert srv.l7.stats.spans_in == 20
        assert srv.l4.stats.flows_in == 20
        client = TestClient(srv.app)
        r = client.post("/v1/query/", json={
            "sql": "SELECT request_resource, Count(*) AS c FROM l7_flow_log "
                   "GROUP BY request_resource ORDER BY c DESC"})
        vals = r.json()["result"]["values"]
        assert sum(v[1] for v in vals) == 20
        assert len(vals) == 4
        r2 = client.post("/v1/query/", json={
            "sql": "SELECT Count(*) AS c FROM l4_flow_log WHERE "
                   "l3_epc_id_0 = 5"})
        assert r2.json()["result"]["values"] == [[20]]
        a.close()
    finally:
        srv.stop()


def test_agent_documents_queryable():
    """Agent Document stream -> flow_metrics pipeline -> SQL."""
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 9,
                         dict_capacity=1 << 10, time_base_s=0)
    a = Agent(vtap_id=4)
    for frame, ts in http_session(CLIENT, SERVER):
        a.packet(frame, ts)
    a.tick(10**9 * 100)
    payload = a.drain(2)
    hdr = framing.FrameHeader(msg_type=framing.MSG_METRICS, agent_id=4)
    assert srv.receiver.handle_frame(framing.encode_frame(hdr, payload))
    assert len(srv.docs.app_rows) >= 1
    r = srv.engine.query(
        "SELECT Sum(request) AS req, Max(rrt_max) AS m FROM application.agent")
    assert r["values"][0][0] == 1
    assert r["values"][0][1] == 4000
    a.close()


def _tcp_exchange(agent_obj, sport, dport, req, resp, t0=10**9):
    from deepflow_amd.agent.packets import eth_ipv4_tcp, SYN, SYNACK, PSH_ACK
    pkts = [
        (eth_ipv4_tcp(CLIENT, SERVER, sport, dport, SYN, 1), t0),
        (eth_ipv4_tcp(SERVER, CLIENT, dport, sport, SYNACK, 2, 2),
         t0 + 1_000_000),
        (eth_ipv4_tcp(CLIENT, SERVER, sport, dport, PSH_ACK, 2, 3, req),
         t0 + 2_000_000),
        (eth_ipv4_tcp(SERVER, CLIENT, dport, sport, PSH_ACK, 3, 2 + len(req),
                      resp), t0 + 5_000_000),
    ]
    for frame, ts in pkts:
        agent_obj.packet(frame, ts)


def test_pgsql(agent):
    import struct
    sql = b"SELECT 1"
    req = b"Q" + struct.pack(">I", 4 + len(sql) + 1) + sql + b"\x00"
    resp = b"T\x00\x00\x00\x06..."
    _tcp_exchange(agent, 51000, 5432, req, resp)
    agent.tick(10**9 * 100)
    l7 = _decode(agent.drain(1), flow_log.APP_PROTO_LOGS_DATA)
    assert len(l7) == 1
    assert l7[0]["base"]["head"]["proto"] == 61
    assert l7[0]["req"]["resource"] == "SELECT ?"  # literals obfuscated


def test_kafka(agent):
    import struct
    cid = b"myclient"
    body = struct.pack(">hhI", 1, 11, 0xBEEF) + \
        struct.pack(">h", len(cid)) + cid + b"restoffetch"
    req = struct.pack(">I", len(body)) + body
    resp_body = struct.pack(">I", 0xBEEF) + b"resp"
    resp = struct.pack(">I", len(resp_body)) + resp_body
    _tcp_exchange(agent, 52000, 9092, req, resp)
    agent.tick(10**9 * 100)
    l7 = _decode(agent.drain(1), flow_log.APP_PROTO_LOGS_DATA)
    assert len(l7) == 1
    assert l7[0]["base"]["head"]["proto"] == 100
    assert l7[0]["req"]["req_type"] == "Fetch"
    assert l7[0]["req"]["domain"] == "myclient"


def test_mongodb(agent):
    import struct
    body = b"\x00" * 10
    req = struct.pack("<iiii", 16 + len(body), 777, 0, 2013) + body
    resp = struct.pack("<iiii", 16 + len(body), 1, 777, 1) + body
    _tcp_exchange(agent, 53000, 27017, req, resp)
    agent.tick(10**9 * 100)
    l7 = _decode(agent.drain(1), flow_log.APP_PROTO_LOGS_DATA)
    assert len(l7) == 1
    assert l7[0]["base"]["head"]["proto"] == 81
    assert l7[0]["req"]["req_type"] == "OP_MSG"


def test_guard_and_sync():
    from fastapi.testclient import TestClient
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 9,
                         dict_capacity=1 << 10)
    from deepflow_amd.store.kg import KgInfo
    srv.controller.update_platform({(7, CLIENT): KgInfo(pod_id=1)})
    client = TestClient(srv.app)
    a = Agent(vtap_id=9)
    assert a.guard_check(max_memory_mb=10**6) == 0
    assert a.guard_check(max_memory_mb=1) & Agent.EXC_MEM_LIMIT

    def post(payload):
        return client.post("/v1/sync/", json=payload).json()

    resp = a.sync_with_controller(post)
    assert "config" in resp and "platform" in resp
    # second sync: versions caught up -> nothing pushed
    resp2 = a.sync_with_controller(post)
    assert "config" not in resp2 and "platform" not in resp2
    # pushed platform applied to the labeler: client ip now labels epc=7
    for frame, ts in http_session(CLIENT, SERVER, sport=47000):
        a.packet(frame, ts)
    a.tick(10**9 * 100)
    l7 = _decode(a.drain(1), flow_log.APP_PROTO_LOGS_DATA)
    assert l7[0]["base"]["l3_epc_id_src"] == 7
    # exception visible to the controller monitor
    a.guard_check(max_memory_mb=1)
    a.sync_with_controller(post)
    agents = client.get("/v1/agents/").json()
    me = next(x for x in agents if x["agent_id"] == 9)
    assert me["exceptions"] & Agent.EXC_MEM_LIMIT
    a.close()


def test_tls_sni(agent):
    import struct
    sni = b"api.internal.example"
    ext = struct.pack(">HHHBH", 0, len(sni) + 5, len(sni) + 3, 0,
                      len(sni)) + sni
    body = (b"\x03\x03" + b"\x00" * 32      # version + random
            + b"\x00"                        # session id len
            + struct.pack(">H", 2) + b"\x13\x01"  # cipher suites
            + b"\x01\x00"                    # compression
            + struct.pack(">H", len(ext)) + ext)
    hs = b"\x01" + struct.pack(">I", len(body))[1:] + body
    rec = b"\x16\x03\x01" + struct.pack(">H", len(hs)) + hs
    server_hello = b"\x16\x03\x03\x00\x06\x02\x00\x00\x02\x03\x03"
    from deepflow_amd.agent.packets import eth_ipv4_tcp, SYN, SYNACK, PSH_ACK
    t0 = 10**9
    pkts = [
        (eth_ipv4_tcp(CLIENT, SERVER, 48000, 443, SYN, 1), t0),
        (eth_ipv4_tcp(SERVER, CLIENT, 443, 48000, SYNACK, 2, 2), t0 + 10**6),
        (eth_ipv4_tcp(CLIENT, SERVER, 48000, 443, PSH_ACK, 2, 3, rec),
         t0 + 2 * 10**6),
        (eth_ipv4_tcp(SERVER, CLIENT, 443, 48000, PSH_ACK, 3, 2 + len(rec),
                      server_hello), t0 + 4 * 10**6),
    ]
    for frame, ts in pkts:
        agent.packet(frame, ts)
    agent.tick(10**9 * 100)
    l7 = _decode(agent.drain(1), flow_log.APP_PROTO_LOGS_DATA)
    assert len(l7) == 1
    assert l7[0]["base"]["head"]["proto"] == 121
    assert l7[0]["req"]["domain"] == "api.internal.example"


def test_compressed_sender():
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 9,
                         dict_capacity=1 << 10, time_base_s=0)
    a = Agent(vtap_id=6)
    for i in range(10):
        for frame, ts in http_session(CLIENT + i, SERVER, sport=45000 + i):
            a.packet(frame, ts)
    a.tick(10**9 * 100)
    payload = a.drain(1)
    framed = a.frame(1, payload, compress=True)
    # encoder byte says zstd and the server decompresses transparently
    assert framed[7] == framing.ENCODER_ZSTD
    assert len(framed) < len(payload)
    assert srv.receiver.handle_frame(framed)
    assert srv.l7.stats.spans_in == 10
    a.close()


def test_pgsql_pipelined(agent):
    """Two 'Q' messages in one segment (PG pipelining): each query gets
    its own record, responses matched FIFO by ReadyForQuery ('Z'), with
    a per-query error status from an 'E' before its 'Z'."""
    import struct

    def q(sql: bytes) -> bytes:
        return b"Q" + struct.pack(">I", 4 + len(sql) + 1) + sql + b"\x00"

    def m(t: bytes, body: bytes = b"") -> bytes:
        return t + struct.pack(">I", 4 + len(body)) + body

    req = q(b"SELECT 1") + q(b"INSERT INTO t VALUES (99)")
    # reply: rows + Z (ok), then error + Z
    resp = m(b"T", b"x") + m(b"Z", b"I") + m(b"E", b"boom") + m(b"Z", b"I")
    _tcp_exchange(agent, 51010, 5432, req, resp)
    agent.tick(10**9 * 100)
    l7 = _decode(agent.drain(1), flow_log.APP_PROTO_LOGS_DATA)
    assert len(l7) == 2
    assert l7[0]["req"]["resource"] == "SELECT ?"
    assert l7[0]["resp"].get("status", 0) == 0
    assert l7[1]["req"]["resource"] == "INSERT INTO t VALUES (?)"
    assert l7[1]["resp"]["status"] == 3


def test_mysql_pipelined_requests(agent):
    """A second COM_QUERY before the first response flushes the first
    request as its own record instead of overwriting it."""
    import struct
    from deepflow_amd.agent.packets import eth_ipv4_tcp

    def com_query(sql: bytes) -> bytes:
        return struct.pack("<I", len(sql) + 1)[:3] + b"\x00\x03" + sql

    t0 = 10**9
    sport, dport = 51020, 3306
    SYN, SYNACK, ACK, PSH_ACK = 0x02, 0x12, 0x10, 0x18
    pkts = [
        (eth_ipv4_tcp(CLIENT, SERVER, sport, dport, SYN, 0, 0), t0),
        (eth_ipv4_tcp(SERVER, CLIENT, dport, sport, SYNACK, 0, 1),
         t0 + 1_000_000),
        (eth_ipv4_tcp(CLIENT, SERVER, sport, dport, ACK, 1, 1),
         t0 + 1_100_000),
        (eth_ipv4_tcp(CLIENT, SERVER, sport, dport, PSH_ACK, 1, 1,
                      com_query(b"SELECT a FROM t1 WHERE x = 5")),
         t0 + 2_000_000),
        (eth_ipv4_tcp(CLIENT, SERVER, sport, dport, PSH_ACK, 40, 1,
                      com_query(b"SELECT b FROM t2 WHERE y = 6")),
         t0 + 3_000_000),
        (eth_ipv4_tcp(SERVER, CLIENT, dport, sport, PSH_ACK, 1, 80,
                      b"\x01\x00\x00\x01\x01"), t0 + 5_000_000),
    ]
    for frame, ts in pkts:
        agent.packet(frame, ts)
    agent.tick(10**9 * 100)
    l7 = _decode(agent.drain(1), flow_log.APP_PROTO_LOGS_DATA)
    assert len(l7) == 2
    res = sorted(r["req"]["resource"] for r in l7)
    assert res == ["SELECT a FROM t1 WHERE x = ?",
                   "SELECT b FROM t2 WHERE y = ?"]
