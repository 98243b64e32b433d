"""Live AF_PACKET capture on loopback: real kernel-delivered frames flow
through the C++ engine and come out as L7 records. Skipped when the
environment denies raw sockets."""
import socket
import time

import pytest

from deepflow_amd.agent import Agent
from deepflow_amd.agent.capture import CaptureDispatcher
from deepflow_amd.wire import pb, flow_log, framing


def _raw_ok():
    try:
        s = socket.socket(socket.AF_PACKET, socket.SOCK_RAW, 0)
        s.close()
        return True
    except (PermissionError, OSError):
        return False


pytestmark = pytest.mark.skipif(not _raw_ok(),
                                reason="AF_PACKET not permitted")


def test_live_http_capture():
    a = Agent(vtap_id=3)
    disp = CaptureDispatcher([a], iface="lo")
    disp.start()
    time.sleep(0.2)
    # real TCP round trip over loopback
    srv = socket.socket()
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)
    port = srv.getsockname()[1]
    cli = socket.create_connection(("127.0.0.1", port))
    conn, _ = srv.accept()
    cli.sendall(b"GET /live/check HTTP/1.1\r\nHost: lo.test\r\n\r\n")
    assert conn.recv(4096).startswith(b"GET /live/check")
    conn.sendall(b"HTTP/1.1 200 OK\r\nContent-Length: 2\r\n\r\nok")
    assert cli.recv(4096).startswith(b"HTTP/1.1 200")
    time.sleep(0.4)
    cli.close()
    conn.close()
    srv.close()
    disp.stop()
    st = disp.stats()
    assert st["packets"] >= 6  # handshake + data + teardown, both dirs
    a.tick(1 << 62)
    recs = [pb.decode(r, flow_log.APP_PROTO_LOGS_DATA)
            for r in framing.iter_records(a.drain(1))]
    ours = [r for r in recs
            if r.get("req", {}).get("resource") == "/live/check"]
    assert ours, [r.get("req") for r in recs]
    assert ours[0]["resp"]["code"] == 200
    assert ours[0]["base"]["head"]["rrt"] > 0
    a.close()
