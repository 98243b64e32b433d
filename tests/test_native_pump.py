"""Native receiver pump (ops/csrc/recv_pump.cpp + ingest/native_pump.py):
socket -> deframe -> zstd -> SPSC pinned ring, consumed zero-copy.
"""
import os
import socket
import time

import numpy as np
import pytest

from deepflow_amd.ingest.native_pump import NativePump, PumpServer
from deepflow_amd.wire import framing
from deepflow_amd.ops import native


def _frame(payload: bytes, zstd: bool = False,
           msg_type: int = framing.MSG_PROTOCOLLOG) -> bytes:
    if zstd:
        import ctypes as ct
        lib = native.cpu()
        src = np.frombuffer(payload, dtype=np.uint8)
        dst = np.zeros(len(payload) * 2 + 1024, dtype=np.uint8)
        n = lib.df_zstd_compress(src.ctypes.data, len(src),
                                 dst.ctypes.data, len(dst), 1)
        assert n > 0
        return framing.encode_frame(
            framing.FrameHeader(msg_type=msg_type,
                                encoder=framing.ENCODER_ZSTD),
            dst[:n].tobytes())
    return framing.encode_frame(framing.FrameHeader(msg_type=msg_type),
                                payload)


def _pump_pair(ring_bytes=1 << 20):
    a, b = socket.socketpair()
    p = NativePump(b, ring_bytes=ring_bytes, pin=False)
    return a, p


def _drain(p, n_expected, timeout=10.0):
    out = []
    t_end = time.time() + timeout
    while len(out) < n_expected and time.time() < t_end:
        v = p.poll()
        if v is None:
            time.sleep(0.001)
            continue
        out.append(bytes(v))
        p.advance()
    return out


def test_pump_raw_and_zstd_frames():
    a, p = _pump_pair()
    try:
        pay1 = b"hello-span-payload" * 10
        pay2 = os.urandom(5000) + b"tail"
        a.sendall(_frame(pay1) + _frame(pay2, zstd=True))
        got = _drain(p, 2)
        assert got == [pay1, pay2]
        st = p.stats()
        assert st["frames"] == 2 and st["bad_frames"] == 0
        assert st["payload_bytes"] == len(pay1) + len(pay2)
    finally:
        a.close()
        p.close()


def test_pump_filters_msg_type():
    a, p = _pump_pair()
    try:
        a.sendall(_frame(b"drop-me", msg_type=framing.MSG_SYSLOG) +
                  _frame(b"keep-me"))
        got = _drain(p, 1)
        assert got == [b"keep-me"]
    finally:
        a.close()
        p.close()


def test_pump_ring_wrap_and_backpressure():
    # ring far smaller than the stream: the producer must block on the
    # consumer (backpressure) and wrap markers must be handled
    a, p = _pump_pair(ring_bytes=1 << 14)  # 16 KB ring
    payloads = [bytes([i % 251]) * (900 + 37 * i) for i in range(64)]
    import threading

    def send():
        for pl in payloads:
            a.sendall(_frame(pl))
        a.shutdown(socket.SHUT_WR)

    t = threading.Thread(target=send)
    t.start()
    try:
        got = _drain(p, len(payloads), timeout=20)
        assert got == payloads
        assert p.stats()["frames"] == len(payloads)
    finally:
        t.join()
        a.close()
        p.close()


def test_pump_desync_drops_connection():
    a, p = _pump_pair()
    try:
        a.sendall(b"\xff\xff\xff\xff garbage that is not a frame")
        t_end = time.time() + 5
        while not p.lib.df_pump_done(p.h) and time.time() < t_end:
            time.sleep(0.01)
        assert p.stats()["bad_frames"] == 1
    finally:
        a.close()
        p.close()


def test_pump_server_end_to_end():
    got = []
    srv = PumpServer(lambda v, meta: got.append(bytes(v)), pin=False,
                     ring_bytes=1 << 20).start()
    try:
        payloads = [b"A" * 100, b"B" * 3000, os.urandom(999)]
        conns = [socket.create_connection(("127.0.0.1", srv.port))
                 for _ in range(2)]
        conns[0].sendall(_frame(payloads[0]) + _frame(payloads[1],
                                                      zstd=True))
        conns[1].sendall(_frame(payloads[2]))
        t_end = time.time() + 10
        while len(got) < 3 and time.time() < t_end:
            time.sleep(0.01)
        assert sorted(got) == sorted(payloads)
        st = srv.stats()
        assert st["frames"] == 3 and st["connections"] == 2
        for c in conns:
            c.close()
    finally:
        srv.stop()


def test_pump_throughput_near_wire_speed():
    """The pump must consume at ~the raw socket rate (the Python
    deframe loop loses ~half of it to interpreter work). Measured
    against a same-host raw-recv baseline so the assertion tracks the
    machine instead of a hard-coded number."""
    import threading

    n_frames, pay = 64, os.urandom(4 << 20)

    def tcp_sender(port, blob, reps):
        s = socket.create_connection(("127.0.0.1", port))
        s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        for _ in range(reps):
            s.sendall(blob)
        s.shutdown(socket.SHUT_WR)
        s.close()

    # baseline: raw recv loop, no framing
    srv = socket.socket()
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    srv.bind(("127.0.0.1", 0))
    srv.listen(1)
    t = threading.Thread(target=tcp_sender,
                         args=(srv.getsockname()[1], pay, n_frames))
    t.start()
    conn, _ = srv.accept()
    t0 = time.perf_counter()
    while conn.recv(1 << 20):
        pass
    base_gbps = n_frames * len(pay) / (time.perf_counter() - t0) / 1e9
    t.join()
    conn.close()
    srv.close()

    # pump: same stream, framed. Best of 3 attempts — scheduler noise
    # on a loaded CI host can halve a single run.
    fr = _frame(pay)
    gbps = 0.0
    for _ in range(3):
        srv = socket.socket()
        srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        srv.bind(("127.0.0.1", 0))
        srv.listen(1)
        t = threading.Thread(target=tcp_sender,
                             args=(srv.getsockname()[1], fr, n_frames))
        t.start()
        conn, _ = srv.accept()
        # ring larger than the whole stream: measure pump speed, not
        # ring backpressure
        p = NativePump(conn, ring_bytes=512 << 20, pin=False)
        t0 = time.perf_counter()
        got = 0
        while got < n_frames:
            v = p.poll()
            if v is None:
                time.sleep(0.0002)  # yield the GIL to the test's sender
                continue
            got += 1
            p.advance()
        dt = time.perf_counter() - t0
        t.join()
        srv.close()
        p.close()
        gbps = max(gbps, n_frames * len(pay) / dt / 1e9)
        if gbps > 0.35 * base_gbps:
            break
    print(f"pump {gbps:.2f} GB/s vs raw recv {base_gbps:.2f} GB/s")
    assert gbps > 0.25 * base_gbps


def test_server_native_pump_data_plane():
    """DeepflowServer(native_pump=True): agent frames land through the
    C++ pump and reach SQL, same as through the Python receiver."""
    from deepflow_amd.server import DeepflowServer
    from deepflow_amd.gen.spans import SpanGenConfig, gen_span_payload

    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12, native_pump=True)
    srv.start()
    try:
        cfg = SpanGenConfig(n=50, seed=3)
        records = gen_span_payload(cfg)
        fr = _frame(records, zstd=True)
        s = socket.create_connection(("127.0.0.1", srv.pump_port))
        s.sendall(fr)
        t_end = time.time() + 10
        while srv.l7.stats.spans_in < 50 and time.time() < t_end:
            time.sleep(0.05)
        s.close()
        assert srv.l7.stats.spans_in == 50
        r = srv.engine.query("SELECT COUNT(1) FROM l7_flow_log")
        assert r["values"][0][0] == 50
    finally:
        srv.stop()


@pytest.mark.gpu
def test_pump_to_gpu_pipeline():
    """Wire -> native pump (pinned ring) -> H2D -> GPU ingest: spans
    land in the device store and are queryable (the bench --path e2e
    fast path, minimally)."""
    import ctypes as ct
    import torch
    from deepflow_amd.gen import SpanGenConfig
    from deepflow_amd.gen.spans import gen_span_payload
    from deepflow_amd.ingest import L7IngestPipeline
    from deepflow_amd.store.kg import (KnowledgeGraphTable,
                                       default_platform)
    from deepflow_amd.ops import native as nat

    cfg = SpanGenConfig(n=5000, seed=11, tag_cardinality=300, n_ips=128)
    kg = KnowledgeGraphTable(capacity_pow2=1 << 12, device="cuda")
    kg.update(default_platform(cfg))
    pipe = L7IngestPipeline(device="cuda", segment_rows=1 << 14, kg=kg,
                            dict_capacity=1 << 15,
                            time_base_s=cfg.base_time_ns // 10**9,
                            defer_harvest=False)
    lib = nat.cpu()
    offs_p = torch.empty(5000, dtype=torch.int32, pin_memory=True)
    lens_p = torch.empty(5000, dtype=torch.int32, pin_memory=True)

    def on_frame(view, meta):
        n = int(lib.df_scan_offsets(
            ct.c_void_p(view.ctypes.data), len(view),
            ct.c_void_p(offs_p.data_ptr()),
            ct.c_void_p(lens_p.data_ptr()), 5000))
        pay = torch.from_numpy(view).to("cuda", non_blocking=True)
        pipe.ingest_device(pay, offs_p[:n].to("cuda", non_blocking=True),
                           lens_p[:n].to("cuda", non_blocking=True),
                           view)
        ev = torch.cuda.Event()
        ev.record()
        return ev

    srv = PumpServer(on_frame, pin=True, ring_bytes=64 << 20).start()
    try:
        payload = gen_span_payload(cfg)
        s = socket.create_connection(("127.0.0.1", srv.port))
        s.sendall(_frame(payload, zstd=True))
        t_end = time.time() + 20
        while pipe.stats.spans_in < 5000 and time.time() < t_end:
            time.sleep(0.02)
        s.close()
        torch.cuda.synchronize()
        assert pipe.stats.spans_in == 5000
        assert srv.stats()["bad_frames"] == 0
        rows = pipe.metrics.rows()
        assert sum(r["request"] for r in rows) == 5000
    finally:
        srv.stop()


def test_server_pump_dispatches_all_types():
    """accept_type=-1 pump: metrics/prometheus/flow frames all route to
    their registered handlers with correct header metadata."""
    from deepflow_amd.server import DeepflowServer
    from deepflow_amd.gen.flows import FlowGenConfig, gen_flow_payload

    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 12, native_pump=True)
    srv.start()
    try:
        l4 = gen_flow_payload(FlowGenConfig(n=10, seed=2))
        fr1 = framing.encode_frame(
            framing.FrameHeader(msg_type=framing.MSG_TAGGEDFLOW,
                                agent_id=42, org_id=1), l4)
        s = socket.create_connection(("127.0.0.1", srv.pump_port))
        s.sendall(fr1)
        t_end = time.time() + 10
        while srv.l4.stats.flows_in < 10 and time.time() < t_end:
            time.sleep(0.05)
        s.close()
        assert srv.l4.stats.flows_in == 10
        r = srv.engine.query("SELECT COUNT(1) FROM l4_flow_log")
        assert r["values"][0][0] == 10
    finally:
        srv.stop()


@pytest.mark.gpu
def test_server_native_pump_gpu_feeder():
    """GPU server + native_pump: L7 frames take the coalescing
    GpuL7Feeder (zero-copy pinned ring -> device ingest) and land in
    SQL; other frame types still dispatch through handlers."""
    from deepflow_amd.server import DeepflowServer
    from deepflow_amd.gen.spans import SpanGenConfig, gen_span_payload

    srv = DeepflowServer(device="cuda", tcp_port=0, segment_rows=1 << 14,
                         dict_capacity=1 << 15, native_pump=True,
                         time_base_s=0)
    srv.start()
    try:
        assert srv._l7_feeder is not None
        cfg = SpanGenConfig(n=4000, seed=13, tag_cardinality=200,
                            n_ips=64, base_time_ns=10**9)
        records = gen_span_payload(cfg)
        fr = _frame(records, zstd=True)
        s = socket.create_connection(("127.0.0.1", srv.pump_port))
        s.sendall(fr)
        t_end = time.time() + 20
        while srv.l7.stats.spans_in < 4000 and time.time() < t_end:
            time.sleep(0.05)
        s.close()
        import torch
        torch.cuda.synchronize()
        assert srv.l7.stats.spans_in == 4000
        r = srv.engine.query("SELECT COUNT(1) FROM l7_flow_log")
        assert r["values"][0][0] == 4000
    finally:
        srv.stop()


def test_pump_frame_larger_than_ring_dropped():
    """A frame whose payload exceeds the ring capacity is drained and
    counted as bad instead of deadlocking the producer."""
    a, p = _pump_pair(ring_bytes=1 << 14)  # 16 KB ring
    try:
        big = b"Z" * (64 << 10)            # 64 KB payload
        a.sendall(_frame(big) + _frame(b"after"))
        got = _drain(p, 1, timeout=10)
        assert got == [b"after"]
        assert p.stats()["bad_frames"] == 1
    finally:
        a.close()
        p.close()


@pytest.mark.gpu
def test_multi_org_isolation_gpu():
    """Per-org pipelines on the GPU: spans sent under different org ids
    land in separate device stores and separate SQL scopes."""
    from deepflow_amd.server import DeepflowServer
    from deepflow_amd.gen.spans import SpanGenConfig, gen_span_payload
    import numpy as np

    srv = DeepflowServer(device="cuda", tcp_port=0, segment_rows=1 << 14,
                         dict_capacity=1 << 15, time_base_s=0)
    payload = np.frombuffer(
        gen_span_payload(SpanGenConfig(n=300, seed=17, n_ips=32)),
        dtype=np.uint8)
    srv._on_l7(framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG,
                                   org_id=1), payload)
    srv._on_l7(framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG,
                                   org_id=7), payload)
    srv._on_l7(framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG,
                                   org_id=7), payload)
    import torch
    torch.cuda.synchronize()
    r1 = srv.engine.query("SELECT COUNT(1) FROM l7_flow_log")
    assert r1["values"][0][0] == 300
    eng7 = srv.org_context(7).engine
    r7 = eng7.query("SELECT COUNT(1) FROM l7_flow_log")
    assert r7["values"][0][0] == 600
    assert srv.org_context(7).l7.segments.device == "cuda"
