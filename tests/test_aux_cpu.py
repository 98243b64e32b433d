"""Aux subsystem tests: throttler, 1m rollups, show tag values, eviction."""
import numpy as np

from deepflow_amd.utils.throttler import SamplingThrottler
from deepflow_amd.query.engine import rollup_rows
from deepflow_amd.store.segment import SegmentSet, L7Segment


def test_throttler_under_limit_passes_all():
    t = SamplingThrottler(limit_per_window=1000)
    keep = t.select(500, now_s=100)
    assert len(keep) == 500
    keep2 = t.select(400, now_s=100)
    assert len(keep2) == 400
    assert t.dropped == 0


def test_throttler_caps_window():
    t = SamplingThrottler(limit_per_window=100)
    total = 0
    for _ in range(10):
        total += len(t.select(50, now_s=7))
    assert total <= 100
    assert t.dropped == 500 - total
    # new window resets the budget but inherits last window's rate estimate
    # (500/s vs budget 100 -> p = 0.2), so ~20% of 50 pass
    kept = len(t.select(50, now_s=8))
    assert 1 <= kept <= 25


def test_throttler_deterministic():
    a = SamplingThrottler(limit_per_window=100, seed=1)
    b = SamplingThrottler(limit_per_window=100, seed=1)
    for w in (5, 5, 6):
        ka = a.select(300, now_s=w)
        kb = b.select(300, now_s=w)
        assert np.array_equal(ka, kb)


def test_rollup_rows():
    rows = [
        {"time": 100, "vtap_id": 1, "l7_protocol": 20, "response_status": 0,
         "server_port": 80, "request": 5, "rrt_max": 9, "rrt_sum": 50},
        {"time": 110, "vtap_id": 1, "l7_protocol": 20, "response_status": 0,
         "server_port": 80, "request": 7, "rrt_max": 20, "rrt_sum": 70},
        {"time": 111, "vtap_id": 2, "l7_protocol": 20, "response_status": 0,
         "server_port": 80, "request": 1, "rrt_max": 1, "rrt_sum": 1},
    ]
    out = rollup_rows(rows, 60)
    assert len(out) == 2
    r1 = next(r for r in out if r["vtap_id"] == 1)
    assert r1["time"] == 60 and r1["request"] == 12
    assert r1["rrt_max"] == 20 and r1["rrt_sum"] == 120


def test_segment_eviction():
    ss = SegmentSet(segment_rows=64, device="cpu", cls=L7Segment,
                    max_bytes=SegmentSet.seg_alloc_bytes(
                        L7Segment(64, "cpu")) * 2 + 1000)
    for _ in range(5):
        seg = ss.tail(64)
        seg.n_rows = 64
    assert len(ss.segments) <= 3
    assert ss.evicted_segments >= 2
    assert ss.evicted_rows == ss.evicted_segments * 64


def test_debug_bus():
    """UDP command bus: stats/store/queues served off the data plane."""
    from deepflow_amd.server import DeepflowServer
    from deepflow_amd.utils.debug_bus import debug_call
    from deepflow_amd.wire import pb, flow_log, framing
    srv = DeepflowServer(device="cpu", tcp_port=0, segment_rows=1 << 10,
                         dict_capacity=1 << 11)
    srv.start()
    try:
        rec = {"base": {"start_time": 10**18, "end_time": 10**18 + 10**6,
                        "flow_id": 1, "vtap_id": 1, "tap_side": 1,
                        "head": {"proto": 20, "msg_type": 2, "rrt": 10},
                        "ip_src": 1, "ip_dst": 2, "port_src": 9,
                        "port_dst": 80, "protocol": 6},
               "req": {"req_type": "GET", "domain": "d", "resource": "/r",
                       "endpoint": "/r"}}
        srv.receiver.handle_frame(framing.encode_frame(
            framing.FrameHeader(msg_type=framing.MSG_PROTOCOLLOG, agent_id=7),
            framing.pack_records([pb.encode(rec,
                                            flow_log.APP_PROTO_LOGS_DATA)])))
        r = debug_call(srv.debug_bus.port, "store")
        assert r["result"]["l7_rows"] == 1
        assert r["result"]["dict_entries"] > 0
        r2 = debug_call(srv.debug_bus.port, "agents")
        assert any(k.startswith("7/") for k in r2["result"])
        r3 = debug_call(srv.debug_bus.port, "nope")
        assert "error" in r3 and "store" in r3["cmds"]
        r4 = debug_call(srv.debug_bus.port, "queues")
        assert "decode_queue" in r4["result"]
    finally:
        srv.stop()
