"""Config system + pcap pipeline tests."""
import struct

from fastapi.testclient import TestClient

from deepflow_amd.utils.config import ServerConfig
from deepflow_amd.ingest.pcap_pipeline import PcapPipeline
from deepflow_amd.server import DeepflowServer
from deepflow_amd.wire import framing


def test_config_defaults_and_merge(tmp_path):
    cfg = ServerConfig.load(path=str(tmp_path / "missing.yaml"))
    assert cfg.get("ingester", "listen-port") == 20033
    f = tmp_path / "server.yaml"
    f.write_text("ingester:\n  listen-port: 30033\n  custom-key: 7\n"
                 "querier:\n  group-capacity: 1024\n")
    cfg2 = ServerConfig.load(path=str(f))
    assert cfg2.get("ingester", "listen-port") == 30033
    assert cfg2.get("ingester", "custom-key") == 7
    assert cfg2.get("ingester", "segment-rows") == 1 << 22  # default kept
    assert cfg2.get("querier", "group-capacity") == 1024


def test_pcap_roundtrip():
    p = PcapPipeline(max_flows=4, max_packets_per_flow=3)
    frame = b"\x00" * 60
    payload = b"".join(
        struct.pack("<QQH", 42, 10**9 + i, len(frame)) + frame
        for i in range(5))
    assert p.ingest_payload(payload) == 5
    blob = p.export_pcap(42)
    assert blob is not None
    magic, = struct.unpack_from("<I", blob, 0)
    assert magic == 0xA1B2C3D4
    # capped at 3 packets/flow
    assert p.stats()["packets"] == 3


def test_pcap_http():
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 9,
                         dict_capacity=1 << 10)
    frame = b"\xaa" * 40
    payload = struct.pack("<QQH", 7, 5 * 10**9, len(frame)) + frame
    srv.receiver.handle_frame(framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_RAW_PCAP), payload))
    client = TestClient(srv.app)
    r = client.get("/v1/pcap/7")
    assert r.status_code == 200 and len(r.content) == 24 + 16 + 40
    assert client.get("/v1/pcap/99").status_code == 404


def test_debug_endpoints():
    from deepflow_amd.gen import SpanGenConfig
    from deepflow_amd.gen.spans import gen_span_payload
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 9,
                         dict_capacity=1 << 10)
    srv.receiver.handle_frame(framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG),
        gen_span_payload(SpanGenConfig(n=20, seed=2, tag_cardinality=5))))
    client = TestClient(srv.app)
    st = client.get("/v1/debug/store").json()
    assert st["l7_rows"] == 20
    assert st["layout_version"] >= 2
    th = client.get("/v1/debug/threads").json()
    assert len(th) >= 1
    sp = client.post("/v1/debug/self-profile").json()
    assert sp["status"] == "skipped"
