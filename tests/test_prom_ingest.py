"""Prometheus remote-write ingest + dfstats self-telemetry loop tests."""
import pytest
from fastapi.testclient import TestClient

from deepflow_amd.server import DeepflowServer
from deepflow_amd.wire import pb, prompb, framing


def mk_write_request(n_series=5, n_samples=4, t0_ms=1_700_000_000_000):
    tss = []
    for s in range(n_series):
        tss.append({
            "labels": [
                {"name": "__name__", "value": "node_cpu_seconds_total"},
                {"name": "instance", "value": f"host-{s}:9100"},
                {"name": "mode", "value": "idle" if s % 2 else "user"},
            ],
            "samples": [{"value": 100.0 * s + i,
                         "timestamp": t0_ms + i * 1000}
                        for i in range(n_samples)],
        })
    return pb.encode({"timeseries": tss}, prompb.WRITE_REQUEST)


@pytest.fixture(scope="module")
def server():
    return DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 9,
                          dict_capacity=1 << 10)


def test_prom_ingest_and_query(server):
    hdr = framing.FrameHeader(msg_type=framing.MSG_PROMETHEUS, agent_id=2)
    frame = framing.encode_frame(hdr, mk_write_request())
    assert server.receiver.handle_frame(frame)
    assert len(server.prom.s_series) == 20
    # SmartEncoding: id-encoded storage beats naive strings
    assert server.prom.stored_bytes() < server.prom.naive_bytes()
    # PromQL over the ingested series
    r = server.promql.instant(
        'node_cpu_seconds_total{mode="user"}', 1_700_000_000 + 3)
    assert len(r["data"]["result"]) == 3  # s = 0, 2, 4
    r2 = server.promql.instant(
        'max(node_cpu_seconds_total) by (instance)', 1_700_000_000 + 3)
    vals = {e["metric"]["instance"]: float(e["value"][1])
            for e in r2["data"]["result"]}
    assert vals["host-4:9100"] == 403.0


def test_dfstats_self_loop(server):
    n = server.ingest_self_stats()
    assert n == 1
    assert len(server.system_rows) > 0
    client = TestClient(server.app)
    r = client.post("/v1/query/", json={
        "sql": "SELECT table, spans_in FROM deepflow_system LIMIT 50"})
    body = r.json()
    assert body["OPT_STATUS"] == "SUCCESS", body
