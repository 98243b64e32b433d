"""CLI smoke test against a live HTTP server (uvicorn in a thread)."""
import socket
import threading
import time

import pytest
from click.testing import CliRunner

from deepflow_amd.cli import cli
from deepflow_amd.gen import SpanGenConfig
from deepflow_amd.gen.spans import gen_span_payload
from deepflow_amd.server import DeepflowServer
from deepflow_amd.wire import framing


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.fixture(scope="module")
def live_server():
    import uvicorn
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12)
    cfg = SpanGenConfig(n=50, seed=4, tag_cardinality=10, n_ips=16)
    srv.receiver.handle_frame(framing.encode_frame(
        framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG),
        gen_span_payload(cfg)))
    port = _free_port()
    config = uvicorn.Config(srv.app, host="127.0.0.1", port=port,
                            log_level="error")
    server = uvicorn.Server(config)
    th = threading.Thread(target=server.run, daemon=True)
    th.start()
    deadline = time.time() + 15
    while time.time() < deadline and not server.started:
        time.sleep(0.05)
    yield f"http://127.0.0.1:{port}"
    server.should_exit = True
    th.join(timeout=5)


def test_cli_query(live_server):
    runner = CliRunner()
    r = runner.invoke(cli, ["--server", live_server, "query",
                            "SELECT Count(*) AS cnt FROM l7_flow_log"])
    assert r.exit_code == 0, r.output
    assert "50" in r.output


def test_cli_tables_and_stats(live_server):
    runner = CliRunner()
    r = runner.invoke(cli, ["--server", live_server, "tables"])
    assert "l7_flow_log" in r.output
    r2 = runner.invoke(cli, ["--server", live_server, "stats"])
    assert r2.exit_code == 0


def test_cli_agent_list(live_server):
    import requests
    requests.post(f"{live_server}/v1/sync/", json={"agent_id": 2})
    runner = CliRunner()
    r = runner.invoke(cli, ["--server", live_server, "agent", "list"])
    assert r.exit_code == 0
    assert "2" in r.output


def test_cli_debug_bus(tmp_path):
    from click.testing import CliRunner
    from deepflow_amd.cli import cli
    from deepflow_amd.utils.debug_bus import DebugBus
    bus = DebugBus()
    bus.register("ping", lambda req: {"pong": True})
    bus.start()
    try:
        r = CliRunner().invoke(cli, ["debug", "ping", "--port",
                                     str(bus.port)])
        assert r.exit_code == 0, r.output
        assert '"pong": true' in r.output
    finally:
        bus.stop()


def test_agent_run_live(tmp_path):
    """`cli agent run` end to end: subprocess agent captures loopback
    traffic with the ring and ships it to a live server; the span shows
    up via SQL."""
    import socket
    import subprocess
    import sys
    import time
    import threading
    from deepflow_amd.server import DeepflowServer
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12, time_base_s=0)
    srv.start()
    try:
        proc = subprocess.Popen(
            [sys.executable, "-m", "deepflow_amd.cli", "agent", "run",
             "--server", f"127.0.0.1:{srv.receiver.tcp_port}",
             "--iface", "lo", "--vtap-id", "9", "--no-ebpf",
             "--flush-interval", "0.3"],
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
        # wait for the agent subprocess (torch import takes seconds)
        # to report its capture is attached before generating traffic
        lines = []
        deadline = time.time() + 30
        while time.time() < deadline:
            line = proc.stdout.readline()
            lines.append(line)
            if "sending to" in line or not line:
                break
        assert any("capture on lo" in ln for ln in lines), lines

        def http_roundtrip():
            # real HTTP round trip on loopback for the agent to capture
            s = socket.socket()
            s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            s.bind(("127.0.0.1", 0))
            s.listen(1)
            port = s.getsockname()[1]
            cli_s = socket.create_connection(("127.0.0.1", port))
            conn, _ = s.accept()
            cli_s.sendall(
                b"GET /cli/run HTTP/1.1\r\nHost: cli.test\r\n\r\n")
            conn.recv(4096)
            conn.sendall(
                b"HTTP/1.1 200 OK\r\nContent-Length: 2\r\n\r\nok")
            cli_s.recv(4096)
            cli_s.close()
            conn.close()
            s.close()

        deadline = time.time() + 20
        found = False
        while time.time() < deadline and not found:
            http_roundtrip()
            time.sleep(0.6)
            r = srv.engine.query(
                "SELECT request_resource FROM l7_flow_log "
                "WHERE request_domain = 'cli.test' LIMIT 5")
            found = bool(r["values"])
        proc.terminate()
        out = "".join(lines) + proc.communicate(timeout=10)[0]
        assert found, out
    finally:
        srv.stop()


def test_cli_domain_config_genesis(live_server):
    runner = CliRunner()
    r = runner.invoke(cli, ["--server", live_server, "domain", "add",
                            "mycloud", "--type", "filereader"])
    assert r.exit_code == 0, r.output
    r = runner.invoke(cli, ["--server", live_server, "domain", "list"])
    assert "mycloud" in r.output
    r = runner.invoke(cli, ["--server", live_server, "agent", "config",
                            "default", "--set", "max_cpus=4"])
    assert r.exit_code == 0, r.output
    r = runner.invoke(cli, ["--server", live_server, "agent", "config",
                            "default"])
    assert '"max_cpus": 4' in r.output
    r = runner.invoke(cli, ["--server", live_server, "genesis"])
    assert r.exit_code == 0, r.output
