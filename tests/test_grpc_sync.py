"""Byte-level trident.Synchronizer gRPC framing: client and server speak
real HTTP/2 + HPACK + gRPC message framing over a TCP socket; the server
dispatches onto ControllerLite's version-gated sync."""
import ipaddress

from deepflow_amd.agent.grpc_client import grpc_sync
from deepflow_amd.control import ControllerLite
from deepflow_amd.control.grpc_server import GrpcSyncServer
from deepflow_amd.wire import pb, trident


def test_grpc_sync_roundtrip():
    from deepflow_amd.store.kg import KgInfo
    ctl = ControllerLite()
    ctl.update_platform({
        (7, int(ipaddress.IPv4Address("10.1.2.3"))): KgInfo(pod_id=1),
        (9, int(ipaddress.IPv4Address("10.1.2.4"))): KgInfo(pod_id=2),
    })
    srv = GrpcSyncServer(ctl)
    srv.start()
    try:
        resp = grpc_sync("127.0.0.1", srv.port, {
            "boot_time": 123,
            "ctrl_ip": "192.168.1.50",
            "ctrl_mac": "02:42:ac:11:00:02",
            "host": "agent-host-1",
            "cpu_num": 8,
            "version_platform_data": 0,
            "exception": 0,
        })
        assert resp.get("status", 0) == trident.STATUS_SUCCESS
        # first sync: config + platform pushed (version 0 behind)
        assert "config" in resp
        assert resp["config"]["enabled"] == 1
        pd = pb.decode(resp["platform_data"], trident.PLATFORM_DATA)
        prefixes = {c["prefix"]: c["epc_id"] for c in pd["cidrs"]}
        assert prefixes["10.1.2.3/32"] == 7
        assert prefixes["10.1.2.4/32"] == 9

        # agent registered under the derived id; version-gated second sync
        assert any(a.hostname == "agent-host-1"
                   for a in ctl.agents.values())
        resp2 = grpc_sync("127.0.0.1", srv.port, {
            "ctrl_ip": "192.168.1.50",
            "ctrl_mac": "02:42:ac:11:00:02",
            "host": "agent-host-1",
            "version_platform_data": resp["version_platform_data"],
        })
        assert "platform_data" not in resp2  # unchanged version: no push
    finally:
        srv.stop()


def test_grpc_unknown_method():
    import socket
    import struct
    from deepflow_amd.control.grpc_server import (
        PREFACE, frame, hpack_encode, HpackDecoder, F_SETTINGS, F_HEADERS,
        F_DATA, FLAG_END_HEADERS, FLAG_END_STREAM, FLAG_ACK)
    ctl = ControllerLite()
    srv = GrpcSyncServer(ctl)
    srv.start()
    try:
        s = socket.create_connection(("127.0.0.1", srv.port), timeout=5)
        s.sendall(PREFACE + frame(F_SETTINGS, 0, 0, b""))
        s.sendall(frame(F_HEADERS, FLAG_END_HEADERS | FLAG_END_STREAM, 1,
                        hpack_encode([(":method", "POST"),
                                      (":path", "/nope/Nope")])))
        dec = HpackDecoder()
        status = None
        while status is None:
            hdr = b""
            while len(hdr) < 9:
                hdr += s.recv(9 - len(hdr))
            ln = (hdr[0] << 16) | (hdr[1] << 8) | hdr[2]
            payload = b""
            while len(payload) < ln:
                payload += s.recv(ln - len(payload))
            if hdr[3] == F_SETTINGS and not hdr[4] & FLAG_ACK:
                s.sendall(frame(F_SETTINGS, FLAG_ACK, 0, b""))
            elif hdr[3] == F_HEADERS:
                hs = dict(dec.decode(payload))
                status = hs.get("grpc-status", status)
        assert status == "12"  # UNIMPLEMENTED
        dec.close()
        s.close()
    finally:
        srv.stop()


def test_grpc_push_stream():
    """Push: the server streams a SyncResponse immediately and another
    when the platform version moves (reference trident.proto Push —
    agents stop polling)."""
    import threading
    import time
    from deepflow_amd.agent.grpc_client import grpc_push
    from deepflow_amd.store.kg import KgInfo
    ctl = ControllerLite()
    srv = GrpcSyncServer(ctl)
    srv.start()
    try:
        got = []

        def consume():
            for msg in grpc_push("127.0.0.1", srv.port,
                                 {"ctrl_ip": "10.0.0.9",
                                  "ctrl_mac": "aa:bb"}, max_msgs=2):
                got.append(msg)
        t = threading.Thread(target=consume, daemon=True)
        t.start()
        time.sleep(0.6)
        assert len(got) == 1                      # initial push
        ctl.update_platform({(7, 0x0A000009): KgInfo(pod_id=42)})
        t.join(timeout=8)
        assert len(got) == 2                      # version-change push
        assert "platform_data" in got[1]
    finally:
        srv.stop()


def test_grpc_upgrade_stream():
    import hashlib
    from deepflow_amd.agent.grpc_client import grpc_upgrade
    ctl = ControllerLite()
    ctl.upgrade_blob = bytes(range(256)) * 1024   # 256 KiB "package"
    srv = GrpcSyncServer(ctl)
    srv.start()
    try:
        blob = grpc_upgrade("127.0.0.1", srv.port, {"ctrl_ip": "10.0.0.1"})
        assert blob == ctl.upgrade_blob
    finally:
        srv.stop()


def test_agent_service_alias():
    """The newer agent.Synchronizer service name maps to the same
    handlers (reference agent.proto mirrors trident.proto)."""
    from deepflow_amd.agent.grpc_client import grpc_stream
    from deepflow_amd.wire import trident as T
    ctl = ControllerLite()
    srv = GrpcSyncServer(ctl)
    srv.start()
    try:
        msgs = list(grpc_stream("127.0.0.1", srv.port,
                                "/agent.Synchronizer/Push",
                                {"ctrl_ip": "1.2.3.4"},
                                T.SYNC_REQUEST, T.SYNC_RESPONSE,
                                max_msgs=1))
        # proto3 zero-skip drops status=SUCCESS; the initial push always
        # carries the config payload (agent versions start at 0)
        assert msgs and "config" in msgs[0]
    finally:
        srv.stop()
