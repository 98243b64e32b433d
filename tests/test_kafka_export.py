"""Kafka exporter against an in-process fake broker that byte-parses the
Produce v2 request (MessageSet v1, CRC-checked) and answers Produce
response v2."""
import json
import socket
import struct
import threading
import zlib

from deepflow_amd.export.kafka_exporter import KafkaExporter


class FakeBroker:
    def __init__(self):
        self.sock = socket.socket()
        self.sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self.sock.bind(("127.0.0.1", 0))
        self.sock.listen(1)
        self.port = self.sock.getsockname()[1]
        self.received = []
        self.thread = threading.Thread(target=self._serve, daemon=True)
        self.thread.start()

    def _recv(self, conn, n):
        buf = b""
        while len(buf) < n:
            chunk = conn.recv(n - len(buf))
            if not chunk:
                raise ConnectionError
            buf += chunk
        return buf

    def _serve(self):
        conn, _ = self.sock.accept()
        while True:
            try:
                (ln,) = struct.unpack(">i", self._recv(conn, 4))
                body = self._recv(conn, ln)
            except (ConnectionError, OSError):
                return
            pos = 0
            api, ver, corr = struct.unpack_from(">hhi", body, pos)
            assert (api, ver) == (0, 2)
            pos += 8
            (cl,) = struct.unpack_from(">h", body, pos)
            pos += 2 + cl
            acks, timeout = struct.unpack_from(">hi", body, pos)
            pos += 6
            (nt,) = struct.unpack_from(">i", body, pos)
            pos += 4
            (tl,) = struct.unpack_from(">h", body, pos)
            pos += 2
            topic = body[pos:pos + tl].decode()
            pos += tl
            (nparts,) = struct.unpack_from(">i", body, pos)
            pos += 4
            part, rss = struct.unpack_from(">ii", body, pos)
            pos += 8
            rs = body[pos:pos + rss]
            # walk the MessageSet, CRC-verify each message
            mp = 0
            while mp < len(rs):
                off, msize = struct.unpack_from(">qi", rs, mp)
                mp += 12
                crc = struct.unpack_from(">I", rs, mp)[0]
                mbody = rs[mp + 4: mp + msize]
                assert zlib.crc32(mbody) & 0xFFFFFFFF == crc
                magic, attrs = mbody[0], mbody[1]
                assert magic == 1
                (ts,) = struct.unpack_from(">q", mbody, 2)
                (klen,) = struct.unpack_from(">i", mbody, 10)
                vp = 14 + (klen if klen > 0 else 0)
                (vlen,) = struct.unpack_from(">i", mbody, vp)
                value = mbody[vp + 4: vp + 4 + vlen]
                self.received.append((topic, json.loads(value)))
                mp += msize
            resp = (struct.pack(">i", corr) +
                    struct.pack(">i", 1) +
                    struct.pack(">h", tl) + topic.encode() +
                    struct.pack(">i", 1) +
                    struct.pack(">ihq", part, 0, 42) +
                    struct.pack(">q", -1) +        # log_append_time
                    struct.pack(">i", 0))          # throttle (v2 tail)
            conn.sendall(struct.pack(">i", len(resp)) + resp)


def test_kafka_export_roundtrip():
    broker = FakeBroker()
    exp = KafkaExporter([("127.0.0.1", broker.port)], topic="deepflow-l7")
    rows = [{"resource": "/api/a", "count": 3},
            {"resource": "/api/b", "count": 5}]
    resp = exp.send_rows(rows, timestamp_ms=1700000000000)
    assert resp["topics"]["deepflow-l7"][0]["error"] == 0
    assert resp["topics"]["deepflow-l7"][0]["offset"] == 42
    assert exp.sent == 2 and exp.errors == 0
    got = [v for (t, v) in broker.received if t == "deepflow-l7"]
    assert got == rows


def test_kafka_export_query():
    from deepflow_amd.gen import SpanGenConfig
    from deepflow_amd.gen.spans import gen_span_payload
    from deepflow_amd.ingest import L7IngestPipeline
    from deepflow_amd.query import QueryEngine
    cfg = SpanGenConfig(n=200, seed=3, tag_cardinality=10, n_ips=16,
                        n_services=2, n_resources=4)
    pipe = L7IngestPipeline(device="cpu", segment_rows=1 << 9,
                            dict_capacity=1 << 10,
                            time_base_s=cfg.base_time_ns // 10**9)
    pipe.ingest_frame_payload(gen_span_payload(cfg))
    eng = QueryEngine(pipe, device="cpu")
    broker = FakeBroker()
    exp = KafkaExporter([("127.0.0.1", broker.port)], topic="agg")
    exp.export_query(eng, "SELECT request_resource, Count(*) AS c "
                          "FROM l7_flow_log GROUP BY request_resource")
    vals = {r["request_resource"]: r["c"]
            for (_, r) in broker.received}
    assert sum(vals.values()) == cfg.n


def test_prometheus_exporter():
    """Prometheus text exposition of the rollup tables (reference
    exporters/prometheus counterpart)."""
    from fastapi.testclient import TestClient
    from deepflow_amd.gen import SpanGenConfig
    from deepflow_amd.gen.spans import gen_span_payload
    from deepflow_amd.server import DeepflowServer
    cfg = SpanGenConfig(n=300, seed=6, tag_cardinality=20, n_ips=16,
                        n_services=4, n_resources=8)
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12,
                         time_base_s=cfg.base_time_ns // 10**9)
    srv.l7.ingest_frame_payload(gen_span_payload(cfg))
    client = TestClient(srv.app)
    r = client.get("/metrics")
    assert r.status_code == 200
    body = r.text
    assert "deepflow_application_request{" in body
    assert 'l7_protocol="' in body
    # values sum to the ingested span count
    total = sum(float(line.rsplit(" ", 2)[1])
                for line in body.splitlines()
                if line.startswith("deepflow_application_request{"))
    assert total == cfg.n
