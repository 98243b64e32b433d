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
    server.prom.samples.flush()
    assert server.prom.samples.n == 20
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


def test_controller_global_label_ids_persist():
    """Prometheus label ids are controller-allocated and survive a
    controller restart (reference: persistent metadb ids served via
    GetPrometheusLabelIDs; round-1 ids were per-shard volatile)."""
    from deepflow_amd.control import ControllerLite
    from deepflow_amd.ingest.prom_pipeline import PromPipeline
    ctl = ControllerLite()
    pipe = PromPipeline(id_allocator=ctl.alloc_prom_ids)
    pipe.ingest_labeled_samples([
        ("http_requests_total", {"job": "api", "code": "200"}, 1000, 5.0),
        ("http_requests_total", {"job": "api", "code": "500"}, 1000, 1.0),
    ])
    mid = pipe.metric_names.to_id["http_requests_total"]
    vid = pipe.label_values.to_id["500"]
    assert mid >= 1 and vid >= 1
    # restart: state roundtrip through the checkpoint dict
    ctl2 = ControllerLite()
    ctl2.load_state_dict(ctl.state_dict())
    pipe2 = PromPipeline(id_allocator=ctl2.alloc_prom_ids)
    pipe2.ingest_labeled_samples([
        ("http_requests_total", {"job": "api", "code": "500"}, 2000, 2.0)])
    assert pipe2.metric_names.to_id["http_requests_total"] == mid
    assert pipe2.label_values.to_id["500"] == vid
    # series still hydrate through the global interners
    series = pipe2.series_for("http_requests_total", [])
    assert series and series[0]["metric"]["job"] == "api"


@pytest.mark.gpu
def test_prom_device_store_gpu():
    """Sample columns live in HBM on a GPU server; series extraction is
    a device mask/gather and matches the CPU result exactly."""
    from deepflow_amd.ingest.prom_pipeline import PromPipeline
    rows = [("rpc_latency", {"svc": f"s{i % 3}"}, 1000 * i, float(i))
            for i in range(500)]
    gpu = PromPipeline(device="cuda")
    cpu = PromPipeline(device="cpu")
    gpu.ingest_labeled_samples(rows)
    cpu.ingest_labeled_samples(rows)
    assert gpu.samples.series.device.type == "cuda"
    for matchers in ([], [("svc", "=", "s1")], [("svc", "=~", "s[02]")]):
        a = gpu.series_for("rpc_latency", matchers)
        b = cpu.series_for("rpc_latency", matchers)
        assert [s["samples"] for s in a] == [s["samples"] for s in b]
        assert [s["metric"] for s in a] == [s["metric"] for s in b]
    assert gpu.stored_bytes() == cpu.stored_bytes()
    assert gpu.naive_bytes() == cpu.naive_bytes()
